"""Process/device runtime — rebuild of the reference's L3b layer
(Theano-MPI ``theanompi/lib/base.py`` ``MPI_GPU_Process`` [pub];
SURVEY.md §2.3).

The reference bound a Theano GPU context and built an intra-node NCCL
clique from an MPI-broadcast clique id.  Here each worker is an OS process
that reads RANK/WORLD_SIZE/MASTER_* from the environment (set by the rule
launcher or by torchrun) and joins a torch.distributed process group:
backend "nccl" (= RCCL over xGMI on ROCm) when running on GPUs, "gloo"
for the CPU plumbing mode.  A second gloo group is always created as the
control plane (EASGD requests, GOSGD gossip, validation reductions that
must not touch GPU streams).
"""

import datetime
import os

import torch
import torch.distributed as dist


def _env_opt(name, default=None):
    return os.environ.get(f"THEANOMPI_AMD_OPT_{name.upper()}", default)


def maybe_resume(model, rec=None):
    """True SGD-state resume: ``rule.init(...,
    resume_dir=..., resume_epoch=N)`` restores per-param
    ``W_<i>_<N>.npy`` weights + ``V_<i>_<N>.npy`` momenta, the pickled
    recorder history, and the LR schedule (by re-running
    ``adjust_hyperp(N)``).  Returns the epoch to start from (0 when not
    resuming)."""
    d = _env_opt("resume_dir", "")
    if not d:
        return 0
    epoch = int(_env_opt("resume_epoch", "0"))
    model.load(d, epoch)
    model.adjust_hyperp(epoch)
    ip = os.path.join(d, "inforec.pkl")
    if rec is not None and os.path.exists(ip):
        rec.load(ip)
    # fast-forward per-epoch data state (shard reshuffle counters) so a
    # resumed run sees the same per-epoch shard order as an
    # uninterrupted one
    if hasattr(model, "reset_iter"):
        for _ in range(epoch):
            model.reset_iter("train")
    return epoch


class GPUProcess:
    """Rank/device binding + communicators for one worker process."""

    def __init__(self):
        self.rank = int(os.environ.get("RANK", "0"))
        self.size = int(os.environ.get("WORLD_SIZE", "1"))
        self.local_rank = int(os.environ.get("LOCAL_RANK", str(self.rank)))
        dev = os.environ.get("THEANOMPI_AMD_DEVICE", "")
        if not dev:
            dev = f"cuda:{self.local_rank}" if torch.cuda.is_available() else "cpu"
        self.device_str = dev
        self.device = None          # torch.device, set by init_device()
        self.on_gpu = dev.startswith("cuda")
        self.verbose = self.rank == 0

    # -- device ---------------------------------------------------------------

    def init_device(self):
        if self.on_gpu:
            idx = int(self.device_str.split(":")[1])
            torch.cuda.set_device(idx)
            self.device = torch.device("cuda", idx)
        else:
            self.device = torch.device("cpu")
        return self.device

    # -- communicators --------------------------------------------------------

    def init_comm(self, timeout_s=300):
        """Create the data-plane group (RCCL on GPU, gloo on CPU) and a
        gloo control-plane group.  Reference: get_intranode_comm() NCCL
        clique + MPI COMM_WORLD [pub lib/base.py]."""
        if self.size == 1 and not dist.is_available():
            return
        backend = "nccl" if self.on_gpu else "gloo"
        if not dist.is_initialized():
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "23456")
            try:
                dist.init_process_group(
                    backend=backend,
                    rank=self.rank,
                    world_size=self.size,
                    timeout=datetime.timedelta(seconds=timeout_s),
                )
            except Exception as e:  # rendezvous failed (e.g. the port
                # picked by the rule launcher was grabbed by a concurrent
                # launch between probe and bind).  Exit code 17 tells
                # rule.wait() to relaunch the job on a fresh port.
                import sys
                print(f"[theanompi_amd rank {self.rank}] rendezvous "
                      f"failed: {e}", file=sys.stderr)
                os._exit(17)
        self.comm = dist.group.WORLD
        # control plane: gloo works for CPU tensors regardless of data plane
        if backend != "gloo":
            self.ctrl = dist.new_group(backend="gloo")
        else:
            self.ctrl = self.comm
        return self.comm

    def barrier(self):
        if dist.is_initialized():
            dist.barrier()

    def finalize(self):
        if dist.is_initialized():
            dist.destroy_process_group()

    def control_store(self):
        """Non-blocking control plane: the rendezvous TCPStore.

        gloo's Work.is_completed() never turns true for pending
        irecv/ibarrier in this torch build, so the async rules (EASGD
        requests, GOSGD push announcements) signal through store keys
        (``check`` is non-blocking) and then run MATCHED blocking
        send/recv pairs on the data plane — nothing ever polls a Work.
        """
        from torch.distributed import distributed_c10d as c10d
        return c10d._get_default_store()

    def broadcast_params(self, model, src=0):
        """Make every rank start from rank `src`'s initialisation
        (the reference relied on identical seeding; an explicit bcast is
        robust).  Uses the data-plane group; refreshes GPU layout copies."""
        if self.size == 1 or not dist.is_initialized():
            return
        for p in model.params:
            dist.broadcast(p.data, src=src)
        if hasattr(model, "refresh_all"):
            model.refresh_all()

    # -- model loading --------------------------------------------------------

    def build_model(self, modelfile=None, modelclass=None, extra_config=None):
        """Reflectively import the model class and instantiate it with the
        config dict of the reference's model contract (SURVEY.md §1 L4->L2)."""
        import importlib
        seed = _env_opt("seed")
        if seed is not None:
            torch.manual_seed(int(seed))
        modelfile = modelfile or os.environ["THEANOMPI_AMD_MODELFILE"]
        modelclass = modelclass or os.environ["THEANOMPI_AMD_MODELCLASS"]
        mod = importlib.import_module(modelfile)
        cls = getattr(mod, modelclass)
        config = {
            "rank": self.rank,
            "size": self.size,
            "device": self.device_str,
            "verbose": self.verbose,
        }
        for key in ("batch_size", "epochs", "dtype", "n_iters", "val_iters",
                    "image_size", "classes"):
            v = _env_opt(key)
            if v is not None:
                config[key] = v
        if extra_config:
            config.update(extra_config)
        return cls(config)
