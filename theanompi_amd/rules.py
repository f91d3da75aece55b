"""Rule / launcher layer.

Rebuilds the reference's L5 layer (Theano-MPI ``theanompi/__init__.py``
BSP/EASGD/GOSGD rule classes [pub]): a rule validates the
device list, spawns one worker process per device, and ``wait()`` joins
them.  The reference shelled out to ``mpirun``; this environment has no
MPI, so the rule execs ``python -m theanompi_amd.<worker>`` directly, one
process per rank, with a torch.distributed TCP rendezvous on 127.0.0.1.

Device strings: ``'cuda0'``/``'cuda:0'``/``'gpu0'`` bind GPU 0; ``'cpu'``
entries select the CPU plumbing mode (gloo backend), the moral equivalent
of the reference's BASELINE config #1.
"""

import os
import signal
import socket
import subprocess
import sys
import time

__all__ = ["BSP", "EASGD", "GOSGD"]


def _free_port():
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _parse_device(dev):
    """'cuda0' / 'cuda:0' / 'gpu0' -> ('cuda', 0); 'cpu' -> ('cpu', -1)."""
    d = dev.strip().lower()
    if d == "cpu":
        return ("cpu", -1)
    for prefix in ("cuda:", "cuda", "gpu:", "gpu", "hip:", "hip"):
        if d.startswith(prefix) and d[len(prefix):].isdigit():
            return ("cuda", int(d[len(prefix):]))
    raise ValueError(f"unrecognised device string: {dev!r}")


class _Rule:
    """Common launcher machinery. Subclasses define worker module(s)."""

    #: module run for every rank (overridden by EASGD which is MPMD)
    worker_module = None
    rule_name = None

    def __init__(self):
        self.procs = []
        self._launched = False
        self._relaunch = None
        self._retries_left = 0

    # -- reference-parity API -------------------------------------------------

    def init(self, devices, modelfile, modelclass, **kwargs):
        """Spawn one worker per device.

        devices    : list of device strings ('cuda0', ..., or 'cpu')
        modelfile  : importable module path holding the model class
        modelclass : class name implementing the model contract
        kwargs     : forwarded to workers as THEANOMPI_AMD_OPT_* env vars
                     (exchange_strategy, sync_type, bucket_mb, epochs, ...)
        """
        if self._launched:
            raise RuntimeError("rule already initialised")
        n = len(devices)
        if n < 1:
            raise ValueError("need at least one device")
        parsed = [_parse_device(d) for d in devices]
        fixed_port = kwargs.pop("master_port", None)
        port = fixed_port or _free_port()
        # _free_port is inherently TOCTOU (the socket is closed before
        # rank 0 binds the rendezvous store); a worker losing the race
        # exits 17 and wait() relaunches everything on a fresh port.
        self._relaunch = (None if fixed_port
                          else (parsed, modelfile, modelclass, dict(kwargs)))
        self._retries_left = 3

        self._spawn(parsed, modelfile, modelclass, kwargs, port)
        self._launched = True

    def _spawn(self, parsed, modelfile, modelclass, kwargs, port):
        n = len(parsed)
        env_common = dict(os.environ)
        env_common.update({
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "WORLD_SIZE": str(n),
            "THEANOMPI_AMD_RULE": self.rule_name,
            "THEANOMPI_AMD_MODELFILE": modelfile,
            "THEANOMPI_AMD_MODELCLASS": modelclass,
        })
        for k, v in kwargs.items():
            env_common[f"THEANOMPI_AMD_OPT_{k.upper()}"] = str(v)

        self.procs = []
        for rank, (kind, idx) in enumerate(parsed):
            env = dict(env_common)
            env["RANK"] = str(rank)
            env["LOCAL_RANK"] = str(max(idx, 0))
            env["THEANOMPI_AMD_DEVICE"] = f"{kind}:{idx}" if kind == "cuda" else "cpu"
            mod = self._module_for_rank(rank, n)
            cmd = [sys.executable, "-u", "-m", mod]
            self.procs.append(subprocess.Popen(cmd, env=env))

    def _maybe_relaunch(self, codes):
        """Exit code 17 = lost the rendezvous-port race: relaunch the
        whole job on a fresh port (bounded retries)."""
        if 17 not in codes or self._relaunch is None \
                or self._retries_left <= 0 or any(c == 0 for c in codes):
            return False
        self._retries_left -= 1
        self.terminate()
        parsed, modelfile, modelclass, kwargs = self._relaunch
        self._spawn(parsed, modelfile, modelclass, kwargs, _free_port())
        return True

    def wait(self, timeout=None):
        """Join all workers; return list of exit codes (reference
        rule.wait()).  A rank that dies while others are wedged cannot
        hang the join: once any rank exits non-zero the rest get a
        grace window (transport errors / the heartbeat watchdog
        normally end them) and are then terminated."""
        deadline = time.time() + timeout if timeout else None
        try:
            while True:
                codes = [p.poll() for p in self.procs]
                if all(c is not None for c in codes):
                    if self._maybe_relaunch(codes):
                        continue
                    return codes
                if any(c is not None and c != 0 for c in codes):
                    grace = time.time() + 15
                    while time.time() < grace and any(
                            p.poll() is None for p in self.procs):
                        time.sleep(0.2)
                    self.terminate()
                    codes = [p.poll() for p in self.procs]
                    if self._maybe_relaunch(codes):
                        continue
                    return codes
                if deadline is not None and time.time() > deadline:
                    raise subprocess.TimeoutExpired(
                        "worker", timeout)
                time.sleep(0.2)
        except Exception:
            self.terminate()
            raise

    def terminate(self):
        for p in self.procs:
            if p.poll() is None:
                p.send_signal(signal.SIGTERM)
        t0 = time.time()
        for p in self.procs:
            while p.poll() is None and time.time() - t0 < 10:
                time.sleep(0.2)
            if p.poll() is None:
                p.kill()

    # -- subclass hooks -------------------------------------------------------

    def _module_for_rank(self, rank, n):
        return self.worker_module


class BSP(_Rule):
    """Bulk-synchronous data parallelism (reference bsp_worker.py [pub])."""
    rule_name = "BSP"
    worker_module = "theanompi_amd.bsp_worker"


class EASGD(_Rule):
    """Asynchronous elastic averaging SGD; rank 0 (devices[0]) is the
    parameter server holding the center variables (reference
    easgd_server.py / easgd_worker.py [pub])."""
    rule_name = "EASGD"

    def _module_for_rank(self, rank, n):
        if n < 2:
            raise ValueError("EASGD needs >= 2 devices (server + workers)")
        return ("theanompi_amd.easgd_server" if rank == 0
                else "theanompi_amd.easgd_worker")


class GOSGD(_Rule):
    """Asynchronous gossip SGD (Blot et al.); reference gosgd_worker.py [pub]."""
    rule_name = "GOSGD"
    worker_module = "theanompi_amd.gosgd_worker"
