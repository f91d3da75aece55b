"""Multi-process rule tests on CPU (gloo, world_size 2-3) — the
reference's BASELINE config #1 plumbing mode.  Each test launches real
worker processes through the rule API, exactly as a user session script
would."""

import os
import subprocess
import sys
import textwrap

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_session(tmp_path, body, env_extra=None, timeout=240):
    script = tmp_path / "session.py"
    script.write_text(textwrap.dedent(body))
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env.setdefault("GLOO_SOCKET_IFNAME", "lo")
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    if env_extra:
        env.update(env_extra)
    res = subprocess.run([sys.executable, str(script)], env=env,
                         capture_output=True, text=True, timeout=timeout)
    assert res.returncode == 0, \
        f"stdout:\n{res.stdout[-4000:]}\nstderr:\n{res.stderr[-4000:]}"
    return res


def test_bsp_two_workers_cpu(tmp_path):
    """2-worker BSP on CPU completes; workers end bit-identical (the
    deterministic grad-sum check of SURVEY.md §7 phase 0)."""
    out = str(tmp_path)
    body = f"""
    import os
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    from theanompi_amd import BSP
    rule = BSP()
    rule.init(devices=['cpu', 'cpu'],
              modelfile='theanompi_amd.models.test_model',
              modelclass='TestModel',
              epochs=1, sync_type='cdd', exchange_strategy='ar',
              dump_dir={out!r})
    codes = rule.wait()
    assert codes == [0, 0], codes
    """
    # patch: workers dump their params when THEANOMPI_AMD_OPT_DUMP_DIR set
    _run_session(tmp_path, body)
    a = np.load(os.path.join(out, "params_rank0.npz"))
    b = np.load(os.path.join(out, "params_rank1.npz"))
    assert set(a.files) == set(b.files)
    for k in a.files:
        np.testing.assert_allclose(a[k], b[k], atol=1e-6)


@pytest.mark.parametrize("strategy,n", [("ar", 2), ("asa", 2),
                                        ("asa_xgmi", 2),
                                        ("asa_xgmi", 3)])
def test_bsp_strategies_cpu(tmp_path, strategy, n):
    # the world-3 asa_xgmi case exercises the non-divisible bucket
    # padding (pad to 4*world) on the explicit all-to-all path
    body = f"""
    import os
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    from theanompi_amd import BSP
    rule = BSP()
    rule.init(devices=['cpu'] * {n},
              modelfile='theanompi_amd.models.test_model',
              modelclass='TestModel',
              epochs=1, sync_type='grad', exchange_strategy={strategy!r})
    assert rule.wait() == [0] * {n}
    """
    _run_session(tmp_path, body)


def test_bsp_avg_mode_cpu(tmp_path):
    body = """
    import os
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    from theanompi_amd import BSP
    rule = BSP()
    rule.init(devices=['cpu', 'cpu'],
              modelfile='theanompi_amd.models.test_model',
              modelclass='TestModel',
              epochs=1, sync_type='avg', exchange_strategy='ar')
    assert rule.wait() == [0, 0]
    """
    _run_session(tmp_path, body)


def test_easgd_cpu(tmp_path):
    body = """
    import os
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    from theanompi_amd import EASGD
    rule = EASGD()
    rule.init(devices=['cpu', 'cpu', 'cpu'],
              modelfile='theanompi_amd.models.test_model',
              modelclass='TestModel',
              tau=2, alpha=0.5, n_total_iters=24)
    assert rule.wait() == [0, 0, 0]
    """
    _run_session(tmp_path, body)


def test_easgd_many_workers_fairness(tmp_path):
    """1 server + 4 workers with tau=1 (every iteration requests an
    exchange): the rotating server poll must serve ALL ranks to
    completion — under the old sorted scan a low rank could starve the
    high ones."""
    body = """
    import os
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    from theanompi_amd import EASGD
    rule = EASGD()
    rule.init(devices=['cpu'] * 5,
              modelfile='theanompi_amd.models.test_model',
              modelclass='TestModel',
              tau=1, alpha=0.5, n_total_iters=40)
    assert rule.wait() == [0] * 5
    """
    _run_session(tmp_path, body, timeout=300)


def test_gosgd_cpu(tmp_path):
    body = """
    import os
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    from theanompi_amd import GOSGD
    rule = GOSGD()
    rule.init(devices=['cpu', 'cpu'],
              modelfile='theanompi_amd.models.test_model',
              modelclass='TestModel',
              epochs=1, p_push=0.5)
    assert rule.wait() == [0, 0]
    """
    _run_session(tmp_path, body)


def test_bsp_overlap_equivalence(tmp_path):
    """Overlapped grad exchange == plain grad exchange (same final
    params), 2 CPU workers over gloo."""
    outs = {}
    for tag, overlap in (("ov", 1), ("plain", 0)):
        d = tmp_path / tag
        d.mkdir()
        body = f"""
        import os
        os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
        from theanompi_amd import BSP
        rule = BSP()
        rule.init(devices=['cpu', 'cpu'],
                  modelfile='theanompi_amd.models.test_model',
                  modelclass='TestModel',
                  epochs=1, sync_type='grad', overlap={overlap},
                  seed=77, dump_dir={str(d)!r})
        assert rule.wait() == [0, 0]
        """
        _run_session(tmp_path, body)
        outs[tag] = np.load(os.path.join(str(d), "params_rank0.npz"))
    for k in outs["ov"].files:
        np.testing.assert_allclose(outs["ov"][k], outs["plain"][k],
                                   atol=1e-5)


def test_strategy_equivalence(tmp_path):
    """'ar' and 'asa' produce identical results (same seeded init/data),
    2 CPU workers."""
    outs = {}
    for strat in ("ar", "asa", "asa_xgmi"):
        d = tmp_path / strat
        d.mkdir()
        body = f"""
        import os
        os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
        from theanompi_amd import BSP
        rule = BSP()
        rule.init(devices=['cpu', 'cpu'],
                  modelfile='theanompi_amd.models.test_model',
                  modelclass='TestModel',
                  epochs=1, sync_type='cdd', exchange_strategy={strat!r},
                  seed=55, dump_dir={str(d)!r})
        assert rule.wait() == [0, 0]
        """
        _run_session(tmp_path, body)
        outs[strat] = np.load(os.path.join(str(d), "params_rank0.npz"))
    for k in outs["ar"].files:
        np.testing.assert_allclose(outs["ar"][k], outs["asa"][k],
                                   atol=2e-5)
        np.testing.assert_allclose(outs["ar"][k], outs["asa_xgmi"][k],
                                   atol=2e-5)


def test_heartbeat_aborts_wedged_job(tmp_path):
    """End-to-end failure detection: one BSP rank wedges mid-epoch
    (process and sockets alive, heartbeat silent — the hung-collective
    analog the transport cannot detect); the watchdog on the OTHER
    rank aborts with exit code 13 instead of hanging forever
    (SURVEY §5.3)."""
    import time
    from theanompi_amd import BSP
    from theanompi_amd.lib.heartbeat import Heartbeat
    rule = BSP()
    rule.init(devices=["cpu", "cpu"],
              modelfile="theanompi_amd.models.test_model",
              modelclass="TestModel", epochs=50, sync_type="cdd",
              exchange_strategy="ar", seed=1, heartbeat=4,
              die_at=3, die_rank=1, die_mode="wedge",
              n_iters=4, val_iters=1)
    # rule.wait() itself must not hang on the wedged rank: rank 0
    # aborts via the watchdog and wait() cleans up the straggler
    codes = rule.wait(timeout=120)
    assert codes[0] == Heartbeat.EXIT_CODE, codes


def test_heartbeat_default_on(tmp_path, monkeypatch):
    """The watchdog is on by default: a wedge aborts cleanly with NO
    heartbeat kwarg.  The default timeout (120 s)
    is shrunk via THEANOMPI_HEARTBEAT_DEFAULT so the test runs fast."""
    from theanompi_amd import BSP
    from theanompi_amd.lib.heartbeat import Heartbeat
    monkeypatch.setenv("THEANOMPI_HEARTBEAT_DEFAULT", "4")
    rule = BSP()
    rule.init(devices=["cpu", "cpu"],
              modelfile="theanompi_amd.models.test_model",
              modelclass="TestModel", epochs=50, sync_type="cdd",
              exchange_strategy="ar", seed=1,
              die_at=3, die_rank=1, die_mode="wedge",
              n_iters=4, val_iters=1)
    codes = rule.wait(timeout=120)
    assert codes[0] == Heartbeat.EXIT_CODE, codes


def test_gosgd_checkpoint(tmp_path):
    """GOSGD rank 0 writes the reference-format checkpoint at the end
    of training (W_<i>_<epoch>.npy per param)."""
    import os
    from theanompi_amd import GOSGD
    rule = GOSGD()
    rule.init(devices=["cpu", "cpu"],
              modelfile="theanompi_amd.models.test_model",
              modelclass="TestModel",
              epochs=1, p_push=0.5, seed=21,
              save_dir=str(tmp_path))
    assert rule.wait() == [0, 0]
    files = sorted(os.listdir(tmp_path))
    assert any(f.startswith("W_0_") and f.endswith(".npy")
               for f in files), files


def test_foreign_module_under_bsp(tmp_path):
    """Reference parity (SURVEY §2.4 Wide-ResNet row): the contract is
    framework-agnostic.  A CNN written with stock torch.nn — no
    BaseModel inheritance, no framework layers — trains under BSP via
    the ContractAdapter, and ranks end bit-identical."""
    out = str(tmp_path)
    body = f"""
    import os
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    from theanompi_amd import BSP
    rule = BSP()
    rule.init(devices=['cpu', 'cpu'],
              modelfile='theanompi_amd.models.module_adapter',
              modelclass='PlainTorchCNN',
              epochs=1, sync_type='cdd', exchange_strategy='ar',
              seed=31, dump_dir={out!r})
    assert rule.wait() == [0, 0]
    """
    _run_session(tmp_path, body)
    a = np.load(os.path.join(out, "params_rank0.npz"))
    b = np.load(os.path.join(out, "params_rank1.npz"))
    for k in a.files:
        np.testing.assert_allclose(a[k], b[k], atol=1e-6)


def test_foreign_module_under_gosgd(tmp_path):
    body = """
    import os
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    from theanompi_amd import GOSGD
    rule = GOSGD()
    rule.init(devices=['cpu', 'cpu'],
              modelfile='theanompi_amd.models.module_adapter',
              modelclass='PlainTorchCNN',
              epochs=1, p_push=0.5, seed=32)
    assert rule.wait() == [0, 0]
    """
    _run_session(tmp_path, body)


def test_gosgd_zoo_scale_vgg16(tmp_path):
    """GOSGD at zoo scale: VGG-16's full 138 M
    params travel per push; 2 ranks, every iteration pushes."""
    body = """
    import os
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    from theanompi_amd import GOSGD
    rule = GOSGD()
    rule.init(devices=['cpu', 'cpu'],
              modelfile='theanompi_amd.models.vgg16',
              modelclass='VGG16',
              epochs=1, p_push=1.0, seed=17,
              batch_size=2, n_iters=2, val_iters=1)
    assert rule.wait() == [0, 0]
    """
    _run_session(tmp_path, body, timeout=600)


def test_resume_bit_equal(tmp_path):
    """resume is true SGD-state resume.  Train 4 epochs
    uninterrupted vs 2 epochs + checkpoint + resume(2) + 2 epochs: the
    final parameters must be bit-equal (weights + momenta + LR schedule
    all restored)."""
    full = tmp_path / "full"
    ck = tmp_path / "ck"
    resumed = tmp_path / "resumed"
    for d in (full, ck, resumed):
        d.mkdir()
    common = ("modelfile='theanompi_amd.models.test_model', "
              "modelclass='TestModel', sync_type='cdd', "
              "exchange_strategy='ar', seed=99")
    body = f"""
    import os
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    from theanompi_amd import BSP
    # run A: 4 uninterrupted epochs
    r = BSP(); r.init(devices=['cpu', 'cpu'], {common},
                      epochs=4, dump_dir={str(full)!r})
    assert r.wait() == [0, 0]
    # run B1: 2 epochs, checkpoint at epoch 2
    r = BSP(); r.init(devices=['cpu', 'cpu'], {common},
                      epochs=2, save_dir={str(ck)!r}, save_freq=2)
    assert r.wait() == [0, 0]
    # run B2: resume from epoch 2, train to 4
    r = BSP(); r.init(devices=['cpu', 'cpu'], {common},
                      epochs=4, resume_dir={str(ck)!r}, resume_epoch=2,
                      dump_dir={str(resumed)!r})
    assert r.wait() == [0, 0]
    """
    _run_session(tmp_path, body, timeout=480)
    a = np.load(os.path.join(str(full), "params_rank0.npz"))
    b = np.load(os.path.join(str(resumed), "params_rank0.npz"))
    assert set(a.files) == set(b.files)
    for k in a.files:
        np.testing.assert_array_equal(a[k], b[k])


@pytest.mark.parametrize("world,port", [(2, 29771), (4, 29773),
                                        (8, 29775)])
def test_bench_multirank_cpu(tmp_path, world, port):
    """The bench contract under torch.distributed.run at world sizes 2
    and 4 (the driver's multi-GPU invocation shape), CPU/gloo — exactly
    ONE JSON line, whole-job aggregate, from rank 0."""
    import json
    import subprocess
    import sys
    import os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(world), "--master-addr", "127.0.0.1",
         "--master-port", str(port), "bench.py", "--gpus", str(world),
         "--model", "cifar10", "--batch", "8", "--steps", "2",
         "--warmup", "1"],
        capture_output=True, text=True, timeout=600, cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines()
             if ln.startswith("{")]
    assert len(lines) == 1, lines          # ONE line, from rank 0
    d = json.loads(lines[0])
    assert d["n_gpus"] == world
    assert d["config"]["parallelism"] == f"dp{world}"
    assert d["config"]["global_batch"] == 8 * world


def test_bench_googlenet_cpu():
    """BASELINE names AlexNet AND GoogLeNet: the GoogLeNet bench config
    must be driver-invokable (VERDICT r1 missing #5).  One tiny CPU
    step through the full inception graph."""
    import json
    import subprocess
    import sys
    import os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "googlenet",
         "--batch", "2", "--steps", "1", "--warmup", "0"],
        capture_output=True, text=True, timeout=600, cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    d = json.loads([ln for ln in out.stdout.splitlines()
                    if ln.startswith("{")][0])
    assert d["config"]["model"] == "googlenet-2b"


def test_foreign_module_under_easgd(tmp_path):
    """Foreign-framework contract under the third rule family too: a
    plain torch nn.Module (module_adapter.PlainTorchCNN) trains as an
    EASGD worker set with the server holding its center variables
    (SURVEY.md §2.4 WRN row; VERDICT r1 missing #4 covered BSP/GOSGD)."""
    body = """
    import os
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    from theanompi_amd import EASGD
    rule = EASGD()
    rule.init(devices=['cpu', 'cpu', 'cpu'],
              modelfile='theanompi_amd.models.module_adapter',
              modelclass='PlainTorchCNN',
              tau=2, alpha=0.5, n_total_iters=12, seed=7)
    assert rule.wait() == [0, 0, 0]
    """
    _run_session(tmp_path, body)


def test_relaunch_on_rendezvous_port_race(tmp_path):
    """rules._maybe_relaunch (the _free_port TOCTOU):
    exit code 17 from every rank relaunches the whole job on a fresh
    port with the ORIGINAL kwargs; success codes or exhausted retries
    do not."""
    body = """
    import os
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    from theanompi_amd import BSP
    rule = BSP()
    rule.init(devices=['cpu', 'cpu'],
              modelfile='theanompi_amd.models.test_model',
              modelclass='TestModel', epochs=1, seed=5)
    assert rule.wait() == [0, 0]

    # negative paths: any success, or no 17, must NOT respawn
    assert not rule._maybe_relaunch([0, 17])
    assert not rule._maybe_relaunch([1, 1])

    # positive path: all-17 respawns with the saved kwargs and the
    # relaunched job runs to completion on the fresh port
    assert rule._retries_left == 3
    assert rule._maybe_relaunch([17, 17])
    assert rule._retries_left == 2
    assert rule.wait() == [0, 0]

    # retries are bounded
    rule._retries_left = 0
    assert not rule._maybe_relaunch([17, 17])

    # a fixed master_port disables the relaunch machinery entirely
    rule2 = BSP()
    rule2.init(devices=['cpu'],
               modelfile='theanompi_amd.models.test_model',
               modelclass='TestModel', epochs=1, seed=5,
               master_port=29771)
    assert rule2.wait() == [0]
    assert not rule2._maybe_relaunch([17])
    """
    _run_session(tmp_path, body, timeout=360)


def test_torchrun_launched_worker(tmp_path):
    """README multi-node contract: the worker modules bootstrap from
    torchrun-style env alone (RANK/LOCAL_RANK/MASTER_* + the
    THEANOMPI_AMD_* config vars), with no Rule launcher in the loop —
    this is the documented path to multi-node."""
    import subprocess
    import sys
    env = dict(os.environ)
    env.update({
        "GLOO_SOCKET_IFNAME": "lo",
        "PYTHONPATH": REPO + os.pathsep + env.get("PYTHONPATH", ""),
        "THEANOMPI_AMD_MODELFILE": "theanompi_amd.models.test_model",
        "THEANOMPI_AMD_MODELCLASS": "TestModel",
        "THEANOMPI_AMD_OPT_EPOCHS": "1",
        "THEANOMPI_AMD_OPT_SEED": "3",
    })
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29613", "-m", "theanompi_amd.bsp_worker"],
        env=env, cwd=str(tmp_path), capture_output=True, text=True,
        timeout=240)
    assert out.returncode == 0, \
        f"stdout:\n{out.stdout[-3000:]}\nstderr:\n{out.stderr[-3000:]}"
    assert "val: cost" in out.stdout


def test_gosgd_merge_invariants():
    """Blot et al. GoSGD algebra at unit level (SURVEY §2.2 GOSGD row):
    the receiver's merge w <- (a*w + a_j*w_j)/(a+a_j), a <- a+a_j must
    (1) be the alpha-weighted mean, (2) conserve total alpha across the
    pair given the sender's halving, (3) conserve the alpha-weighted
    parameter sum — the consensus invariant that makes the gossip
    average converge to the true average."""
    import torch
    from theanompi_amd.lib.exchanger import GOSGD_Exchanger

    w_r = torch.nn.Parameter(torch.tensor([2.0, -1.0]))
    recv = GOSGD_Exchanger({}, [w_r])
    assert recv.alpha == 1.0

    # sender side (emulated): alpha 1.0 halves to 0.5, wire carries
    # [alpha_j, w_j...]
    a_j = 0.5
    w_j = torch.tensor([4.0, 3.0])
    buf = torch.empty(3, dtype=torch.float32)
    buf[0] = a_j
    buf[1:] = w_j

    # pair state before the push: both ranks alpha=1; sender holds w_j
    weighted_before = 1.0 * w_r.detach().clone() + 1.0 * w_j

    recv.try_merge_buf(buf)

    want = (1.0 * torch.tensor([2.0, -1.0]) + a_j * w_j) / (1.0 + a_j)
    assert torch.allclose(w_r.detach(), want)
    assert recv.alpha == 1.5
    # alpha conservation: receiver 1.5 + sender's retained 0.5 == 2.0
    assert recv.alpha + 0.5 == 2.0
    # alpha-weighted parameter sum over the pair is conserved
    weighted_after = recv.alpha * w_r.detach() + 0.5 * w_j
    assert torch.allclose(weighted_after, weighted_before)
