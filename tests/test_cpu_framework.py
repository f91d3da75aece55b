"""CPU plumbing tests: model contract, optimizer, recorder, checkpoints,
conv-plan table construction.  (Multi-process rule tests live in
test_rules_cpu.py; GPU kernel numerics in test_kernels_gpu.py.)"""

import os

import numpy as np
import pytest
import torch

from theanompi_amd.lib.exchanger_strategy import flatten_params
from theanompi_amd.lib.opt import ParamRec, SGDMomentum
from theanompi_amd.lib.recorder import Recorder
from theanompi_amd.lib import helper_funcs
from theanompi_amd.models.layers import ConvPlan, PIECE
from theanompi_amd.models.test_model import TestModel
from theanompi_amd.ops.functional import cpad_for


def make_model(**kw):
    cfg = {"rank": 0, "size": 1, "device": "cpu", "verbose": False}
    cfg.update(kw)
    m = TestModel(cfg)
    m.compile_iter_fns(kw.get("sync_type", "cdd"))
    return m


def test_cpad_for():
    assert cpad_for(3) == 4
    assert cpad_for(16) == 16
    assert cpad_for(17) == 32
    assert cpad_for(48) == 48
    assert cpad_for(96) == 96


def test_model_contract_surface():
    m = make_model()
    for attr in ("params", "vels", "data", "batch_size"):
        assert hasattr(m, attr)
    for meth in ("compile_iter_fns", "train_iter", "val_iter",
                 "adjust_hyperp", "load", "save", "reset_iter"):
        assert callable(getattr(m, meth))
    assert len(m.params) == len(m.vels) > 0


def test_train_loss_decreases():
    torch.manual_seed(0)
    m = make_model(sync_type="grad")
    losses = []
    for i in range(30):
        loss, err = m.train_iter(i % m.data.n_batch_train)
        m.apply_update()
        losses.append(float(loss))
    assert losses[-1] < losses[0], losses


def test_cdd_equals_plain_sgd():
    """cdd phase1+phase2 on one worker == full fused step (reference
    semantics: with N=1 the exchanged vels are the local vels)."""
    torch.manual_seed(1)
    m1 = make_model(sync_type="cdd")
    torch.manual_seed(1)
    m2 = make_model(sync_type="grad")
    for p1, p2 in zip(m1.params, m2.params):
        assert torch.allclose(p1, p2)
    for i in range(3):
        m1.train_iter(i)
        m1.apply_update()          # phase2 applies vels
        m2.train_iter(i)
        m2.apply_update()          # full step from grads
    for p1, p2 in zip(m1.params, m2.params):
        assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_val_iter():
    m = make_model()
    cost, top1, top5 = m.val_iter(0)
    assert np.isfinite(cost)
    assert 0.0 <= top1 <= 1.0 and 0.0 <= top5 <= top1 + 1e-9


def test_adjust_hyperp():
    m = make_model()
    lr0 = m.opt.lr
    m.adjust_hyperp(1)
    assert m.opt.lr == pytest.approx(lr0 * 0.1)


def test_checkpoint_roundtrip(tmp_path):
    torch.manual_seed(2)
    m = make_model(sync_type="grad")
    for i in range(2):
        m.train_iter(i)
        m.apply_update()
    m.save(str(tmp_path), epoch=1)
    # files follow the reference naming pattern
    assert (tmp_path / "W_0_1.npy").exists()
    assert (tmp_path / "V_0_1.npy").exists()
    ref = [p.detach().clone() for p in m.params]
    refv = [v.clone() for v in m.vels]
    torch.manual_seed(99)
    m2 = make_model(sync_type="grad")
    m2.load(str(tmp_path), epoch=1)
    for a, b in zip(ref, m2.params):
        assert torch.allclose(a, b, atol=1e-6)
    for a, b in zip(refv, m2.vels):
        assert torch.allclose(a, b, atol=1e-6)


def test_recorder_roundtrip(tmp_path):
    r = Recorder({"rank": 0, "verbose": False})
    r.start()
    r.end("calc")
    r.train_error(1, 2.0, 0.5)
    r.print_train_info(0, force=True)
    r.val_error(1, 1.5, 0.4, 0.1)
    p = tmp_path / "inforec.pkl"
    r.save(str(p))
    r2 = Recorder({"verbose": False}).load(str(p))
    assert r2.train_info == r.train_info
    assert r2.val_info == r.val_info


def test_flatten_params():
    ts = [torch.randn(7), torch.randn(3, 5), torch.randn(11)]
    flat, views = flatten_params(ts, bucket_bytes=1 << 30)
    assert len(flat) == 1
    for v, t in zip(views, ts):
        v.copy_(t)
    off = 0
    for t in ts:
        assert torch.equal(flat[0][off:off + t.numel()], t.reshape(-1))
        off += t.numel()


def test_flatten_params_bucketing():
    ts = [torch.randn(1000) for _ in range(10)]
    flat, views = flatten_params(ts, bucket_bytes=4 * 2048)
    assert len(flat) > 1
    assert sum(f.numel() for f in flat) == 10 * 1000


def test_sgd_momentum_semantics():
    w = torch.zeros(4)
    w.grad = torch.ones(4)
    rec = ParamRec(w)
    opt = SGDMomentum([rec], lr=0.1, momentum=0.9, weight_decay=0.0)
    opt.step()
    assert torch.allclose(w, torch.full((4,), -0.1))
    w.grad = torch.ones(4)
    opt.step()
    # v = 0.9*(-0.1) - 0.1 = -0.19 ; w = -0.1 - 0.19 = -0.29
    assert torch.allclose(w, torch.full((4,), -0.29))


# ---------------------------------------------------------------------------
# conv plan geometry (pure host logic -> CPU-testable)
# ---------------------------------------------------------------------------

def test_convplan_table_dense():
    # 3x3 conv, C=32 (one combined span per r of length 96)
    plan = ConvPlan(2, 8, 8, 32, 64, 3, 3, (1, 1), (1, 1), 1, "cpu")
    assert plan.Kdf == 320          # 3*96 = 288, rounded up to KPAD=64
    tab = plan.table.numpy()
    # piece j inside row r covers elements r*Wp*C + j16
    Wp = 10
    for j, off in enumerate(tab):
        if j * PIECE >= 288:
            assert off == 0          # KPAD padding entries
            continue
        r = (j * PIECE) // 96
        pos = (j * PIECE) % 96
        assert off == r * Wp * 32 + pos


def test_convplan_table_stem():
    # 11x11 stride-4 stem, C=4: runs of 44, padded to 48
    plan = ConvPlan(1, 227, 227, 4, 96, 11, 11, (4, 4), (0, 0), 1, "cpu")
    assert plan.Kdf == (11 * 48 + 63) // 64 * 64
    tab = plan.table.numpy()
    Wp = 227
    for j, off in enumerate(tab):
        if j * PIECE >= 11 * 48:
            assert off == 0          # KPAD padding entries
            continue
        r = (j * PIECE) // 48
        pos = (j * PIECE) % 48
        if pos < 44:
            assert off == r * Wp * 4 + pos
        else:
            assert off == r * Wp * 4  # padding piece points at span start


def test_convplan_grouped():
    # grouped conv: spans per (r, s) of length Cg
    plan = ConvPlan(1, 27, 27, 96, 256, 5, 5, (1, 1), (2, 2), 2, "cpu")
    assert plan.Cg == 48
    assert plan.Kdf == (5 * 5 * 48 + 63) // 64 * 64
    assert plan.Kgp == 128
    assert plan.Kphys == 256


def test_convplan_dgrad_classes():
    # stride-2 conv -> 4 dgrad classes
    plan = ConvPlan(1, 8, 8, 16, 16, 3, 3, (2, 2), (1, 1), 1, "cpu")
    classes = [p for p in plan.dgrad_plans if p is not None]
    assert len(classes) == 4
    for p in classes:
        assert p["Kdf2"] % PIECE == 0


def test_convplan_flat_roundtrip():
    plan = ConvPlan(1, 8, 8, 16, 32, 3, 3, (1, 1), (1, 1), 1, "cpu")
    w = torch.randn(32, 16, 3, 3)
    flat = plan.w_to_flat(w)
    back = plan.flat_to_w(flat)
    assert torch.allclose(w, back)


def test_convplan_flat_roundtrip_grouped():
    plan = ConvPlan(1, 8, 8, 32, 24, 3, 3, (1, 1), (1, 1), 2, "cpu")
    w = torch.randn(24, 16, 3, 3)     # Kg=12 per group, Cg=16
    flat = plan.w_to_flat(w)
    assert flat.shape == (2, plan.Kgp, plan.Kdf)
    back = plan.flat_to_w(flat)
    assert torch.allclose(w, back)


def test_imagenet_shard_loader(tmp_path):
    """Shard dataset + separate loader child process."""
    from theanompi_amd.models.data.imagenet import (ImageNetData,
                                                    write_synthetic_shards)
    paths = write_synthetic_shards(str(tmp_path), n_shards=3,
                                   images_per_shard=8, hw=40)
    data = ImageNetData(paths, paths[:1], batch_size=8, crop=32,
                        n_classes=1000, device="cpu")
    try:
        assert data.n_batch_train == 3
        x, y = data.get_train_batch(0)
        assert x.shape == (8, 32, 32, 3)
        assert x.dtype == torch.float32
        assert y.shape == (8,)
        xv, yv = data.get_val_batch(0)
        assert xv.shape == (8, 32, 32, 3)
    finally:
        data.close()


def test_imagenet_sub_batches_per_shard(tmp_path):
    """Reference file_batch_size semantics: a shard
    bigger than batch_size feeds multiple iterations — no images are
    silently dropped, and labels line up with their sub-batch."""
    from theanompi_amd.models.data.imagenet import (ImageNetData,
                                                    write_synthetic_shards)
    paths = write_synthetic_shards(str(tmp_path), n_shards=2,
                                   images_per_shard=16, hw=40)
    d = ImageNetData(paths, paths[:1], batch_size=4, crop=40,
                     n_classes=1000, device="cpu", shuffle=False)
    try:
        assert d.sub_batches == 4
        assert d.n_batch_train == 8          # 2 shards x 4 sub-batches
        assert d.n_batch_val == 4
        seen = []
        for i in range(d.n_batch_train):
            x, y = d.get_train_batch(i)
            assert x.shape == (4, 40, 40, 3)
            seen.append(y.clone())
        labels = torch.cat(seen)
        # every training image's label was served exactly once
        import numpy as np
        want = np.concatenate(
            [np.load(p.replace(".npy", "_labels.npy"))
             for p in d.my_shards])
        assert sorted(labels.tolist()) == sorted(want.tolist())
        xv, yv = d.get_val_batch(3)          # last val sub-batch
        assert xv.shape == (4, 40, 40, 3)
    finally:
        d.close()


def test_imagenet_underfilled_shard_tail_zeroed(tmp_path):
    """A shard smaller than the slot leaves no stale images/labels."""
    import numpy as np
    from theanompi_amd.models.data.imagenet import (ImageNetData,
                                                    write_synthetic_shards)
    paths = write_synthetic_shards(str(tmp_path), n_shards=1,
                                   images_per_shard=8, hw=40)
    # a second shard with only 5 images (tail of 3 must be zeroed)
    small = str(tmp_path / "small.npy")
    np.save(small, np.random.RandomState(7).randint(
        0, 256, (5, 40, 40, 3), dtype=np.uint8))
    np.save(small.replace(".npy", "_labels.npy"),
            np.arange(5) + 100)
    d = ImageNetData([paths[0], small], [paths[0]], batch_size=8,
                     crop=32, n_classes=1000, device="cpu", shuffle=False)
    try:
        x0, y0 = d.get_train_batch(0)
        x1, y1 = d.get_train_batch(1)        # the underfilled shard
        assert (y1[:5] >= 100).all()
        assert (y1[5:] == 0).all()           # not shard 0's labels
    finally:
        d.close()


def test_avg_mode_defaults_to_fp32_wire():
    """'avg' ships fp32 master weights — bf16 wire would
    erase small updates.  The worker must default avg to rccl32."""
    from theanompi_amd.bsp_worker import default_strategy
    assert default_strategy("avg", True) == "rccl32"
    assert default_strategy("cdd", True) == "rccl_bf16"
    assert default_strategy("grad", True) == "rccl_bf16"
    assert default_strategy("avg", False) == "ar"


def test_cifar10_data_object():
    from theanompi_amd.models.data.cifar10_data import Cifar10Data
    d = Cifar10Data(batch_size=16, device="cpu", synthetic_train=64,
                    synthetic_val=32)
    x, y = d.get_train_batch(0)
    assert x.shape == (16, 32, 32, 3)
    assert d.n_batch_train == 4


def test_exchanger_per_tensor_mode():
    """Reference parity: tensor-by-tensor exchange."""
    import torch.distributed as dist
    from theanompi_amd.lib.exchanger import BSP_Exchanger
    ps = [torch.ones(5), torch.full((3, 3), 2.0)]
    ex = BSP_Exchanger({"sync_type": "avg", "size": 1,
                        "per_tensor": True}, ps, None)
    ex.exchange()            # size 1: no-op, but exercises the path
    assert torch.allclose(ps[0], torch.ones(5))


def test_strategy_aliases():
    from theanompi_amd.lib.exchanger_strategy import make_strategy
    for name in ("ar", "nccl32", "rccl32", "nccl16", "rccl_bf16",
                 "asa", "asa32", "asa16", "asa_bf16", "asa_xgmi",
                 "asa_xgmi_bf16"):
        assert make_strategy(name) is not None
    import pytest as _pt
    with _pt.raises(ValueError):
        make_strategy("bogus")


def test_imagenet_epoch_reshuffle(tmp_path):
    """reset('train') advances the epoch and reshuffles the shard order
    (per-epoch shard permutation)."""
    from theanompi_amd.models.data.imagenet import (ImageNetData,
                                                    write_synthetic_shards)
    paths = write_synthetic_shards(str(tmp_path), n_shards=6,
                                   images_per_shard=4, hw=40)
    d = ImageNetData(paths, paths[:1], batch_size=4, crop=32,
                     n_classes=10, device="cpu", shuffle=True)
    try:
        order0 = list(d.my_shards)
        d.reset("train")
        order1 = list(d.my_shards)
        assert sorted(order0) == sorted(order1)     # same shard set
        assert order0 != order1                     # new permutation
        x, y = d.get_train_batch(0)                 # still serves batches
        assert x.shape[0] == 4
    finally:
        d.close()


class _DictStore:
    def __init__(self):
        self.d = {}

    def set(self, k, v):
        self.d[k] = v

    def get(self, k):
        return self.d[k]

    def check(self, keys):
        return all(k in self.d for k in keys)


def test_heartbeat_detects_dead_rank():
    """A peer whose counter stops advancing trips the watchdog within
    the timeout; live peers never do."""
    import time
    from theanompi_amd.lib.heartbeat import Heartbeat
    store = _DictStore()
    dead = []
    a = Heartbeat(store, 0, 2, timeout=1.0, interval=0.1,
                  on_dead=dead.append)
    b = Heartbeat(store, 1, 2, timeout=1.0, interval=0.1,
                  on_dead=lambda r: dead.append(("b", r)))
    a.start()
    b.start()
    time.sleep(0.8)
    assert not dead                    # both beating -> no trigger
    b.stop()                           # rank 1 "dies"
    t0 = time.time()
    while not dead and time.time() - t0 < 5:
        time.sleep(0.05)
    a.stop()
    assert dead == [1]                 # a saw rank 1 go silent


def test_heartbeat_no_false_abort_in_bsp(tmp_path):
    """heartbeat=<timeout> as a rule kwarg reaches the workers and a
    healthy 2-rank BSP run completes normally."""
    from theanompi_amd import BSP
    rule = BSP()
    rule.init(devices=["cpu", "cpu"],
              modelfile="theanompi_amd.models.test_model",
              modelclass="TestModel", epochs=1, sync_type="cdd",
              exchange_strategy="ar", seed=11, heartbeat=20,
              dump_dir=str(tmp_path))
    codes = rule.wait()
    assert all(c == 0 for c in codes)


def test_imagenet_mean_image(tmp_path):
    """Dataset mean-image subtraction ((x - mean)/128, the reference's
    img_mean path) on the CPU branch."""
    import numpy as np
    from theanompi_amd.models.data.imagenet import (ImageNetData,
                                                    write_synthetic_shards)
    paths = write_synthetic_shards(str(tmp_path), n_shards=2,
                                   images_per_shard=4, hw=40)
    mean = np.full((32, 32, 3), 100.0, dtype=np.float32)
    mp = str(tmp_path / "mean.npy")
    np.save(mp, mean)
    d = ImageNetData(paths, paths[:1], batch_size=4, crop=32,
                     n_classes=10, device="cpu", shuffle=False,
                     mean_path=mp)
    d2 = ImageNetData(paths, paths[:1], batch_size=4, crop=32,
                      n_classes=10, device="cpu", shuffle=False)
    try:
        xm, _ = d.get_val_batch(0)
        x0, _ = d2.get_val_batch(0)
        # (u8 - 100)/128  vs  u8/128 - 1: differ by (128-100)/128
        assert torch.allclose(xm - x0, torch.full_like(xm, 28.0 / 128.0),
                              atol=1e-5)
    finally:
        d.close()
        d2.close()


def test_row_strided_detector():
    """_row_strided: channel slices of a concat buffer qualify; other
    non-contiguous layouts do not."""
    from theanompi_amd.models.layers import _row_strided
    t = torch.zeros(2, 4, 4, 32)
    assert _row_strided(t)                # contiguous is ld == C
    sl = t[..., 8:24]
    assert _row_strided(sl)
    assert not _row_strided(t.permute(0, 3, 1, 2))
    assert not _row_strided(t[:, ::2])    # strided H: rows not uniform


def test_arena_alias_not_view():
    """_arena_alias shares storage but carries no autograd view
    metadata (so grad accumulation can steal it without a clone)."""
    from theanompi_amd.models.layers import _arena_alias
    arena = torch.zeros(100)
    v = arena[10:30].view(4, 5)
    a = _arena_alias(v)
    assert not a._is_view()
    a[0, 0] = 7.0
    assert arena[10].item() == 7.0
    assert a.shape == (4, 5)


def test_grad_arena_wiring_cpu():
    """On CPU there is no arena (GPU-only fast path) and models still
    train; the opt tolerates grad_arena=None."""
    from theanompi_amd.models.test_model import TestModel
    m = TestModel({"rank": 0, "size": 1, "device": "cpu",
                   "verbose": False, "batch_size": 8, "n_iters": 2,
                   "val_iters": 1})
    assert m.grad_arena is None
    m.compile_iter_fns("grad")
    m.train_iter(0)
    m.apply_update()


def test_alexnet_on_shard_pipeline(tmp_path):
    """AlexNet consumes the real shard pipeline when the session passes
    train_shards (the reference's ImageNet workflow)."""
    from theanompi_amd.models.alex_net import AlexNet
    from theanompi_amd.models.data.imagenet import (ImageNetData,
                                                    write_synthetic_shards)
    paths = write_synthetic_shards(str(tmp_path), n_shards=3,
                                   images_per_shard=4, hw=232)
    m = AlexNet({"rank": 0, "size": 1, "device": "cpu", "verbose": False,
                 "batch_size": 4, "train_shards": ",".join(paths),
                 "val_shards": paths[0]})
    try:
        assert isinstance(m.data, ImageNetData)
        m.compile_iter_fns("grad")
        loss, err = m.train_iter(0)
        m.apply_update()
        assert 0 <= float(err) <= 1
    finally:
        m.data.close()


def test_recorder_jsonl(tmp_path):
    """dump_jsonl emits one machine-readable row per train record with
    the calc/comm/wait split (the scaling-curve harness input)."""
    import json
    from theanompi_amd.lib.recorder import Recorder
    r = Recorder({"rank": 2, "verbose": False})
    for i in range(3):
        r.start()
        r.end("calc")
        r.start()
        r.end("comm")
        r.train_error(i, 2.0 - i, 0.5)
        r.print_train_info(i, force=True)
    p = tmp_path / "rec.jsonl"
    r.dump_jsonl(str(p))
    rows = [json.loads(line) for line in open(p)]
    assert len(rows) == 3
    for row in rows:
        assert set(row) == {"epoch", "iter", "cost", "error", "calc_s",
                            "comm_s", "wait_s", "rank"}
        assert row["rank"] == 2
        assert row["calc_s"] >= 0 and row["comm_s"] >= 0


def test_heartbeat_disabled_paths():
    """maybe_start returns None when disabled (no timeout / world 1);
    Heartbeat.start is a no-op for size < 2."""
    from theanompi_amd.lib.heartbeat import Heartbeat
    hb = Heartbeat(_DictStore(), 0, 1, timeout=5).start()
    assert hb._thread is None          # size 1: nothing to watch
    hb2 = Heartbeat(_DictStore(), 0, 4, timeout=0).start()
    assert hb2._thread is None         # timeout 0: disabled
    hb.stop()
    hb2.stop()


def test_conv_stats_cpu_noop():
    """collect_stats is a GPU-epilogue feature; on CPU the layer still
    works and pop_stats reports nothing."""
    from theanompi_amd.models import layers as L
    conv = L.Conv2d((8, 8, 3), 16, 3, pad=1, device="cpu", batch=2,
                    collect_stats=True)
    x = torch.randn(2, 8, 8, 3)
    y = conv(x)
    assert y.shape == (2, 8, 8, 16)
    assert conv.pop_stats() is None


def test_bench_contract_cpu(tmp_path):
    """bench.py (the driver contract) runs on CPU and prints one valid
    JSON line with the required fields."""
    import json
    import subprocess
    import sys
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "cifar10", "--batch",
         "8", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=300,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for k in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
              "ms_per_step", "higher_is_better", "scaling",
              "vs_baseline", "dtype", "data", "config"):
        assert k in d, k
    assert d["n_gpus"] == 1 and d["steps"] == 2
    assert d["data"] == "synthetic"
    assert d["config"]["global_batch"] == 8


def test_imagenet_augmentation_semantics(tmp_path):
    """SURVEY §3.5 augmentation contract: train batches are randomly
    cropped/mirrored per shard visit (different across epochs), the
    val path is a DETERMINISTIC center crop with no mirror."""
    import numpy as np
    from theanompi_amd.models.data.imagenet import (ImageNetData,
                                                    write_synthetic_shards)
    paths = write_synthetic_shards(str(tmp_path), n_shards=1,
                                   images_per_shard=4, hw=40)
    d = ImageNetData(paths, paths, batch_size=4, crop=32,
                     n_classes=1000, device="cpu", shuffle=False)
    try:
        # val: bit-identical across calls and equal to the manual
        # center crop of the raw uint8 shard
        xv1, yv1 = d.get_val_batch(0)
        xv2, yv2 = d.get_val_batch(0)
        assert torch.equal(xv1, xv2) and torch.equal(yv1, yv2)
        raw = np.load(paths[0])
        o = (40 - 32) // 2
        want = torch.from_numpy(
            raw[:, o:o + 32, o:o + 32, :].astype(np.float32)) \
            * (1 / 128.0) - 1.0
        assert torch.equal(xv1, want)

        # train: the random crop/mirror must stay inside the source
        # image (the batch is a translated window of raw pixel values)
        # and vary across epoch visits with overwhelming probability
        seen = set()
        for epoch in range(6):
            d.reset("train")
            x, _ = d.get_train_batch(0)
            assert x.shape == (4, 32, 32, 3)
            # every served pixel value exists in the raw shard (crop
            # never reads out of bounds / stale slot memory)
            lo = raw.min() / 128.0 - 1.0
            hi = raw.max() / 128.0 - 1.0
            assert x.min() >= lo - 1e-6 and x.max() <= hi + 1e-6
            seen.add(x.numpy().tobytes())
        assert len(seen) > 1, "train crops never varied"
    finally:
        d.close()
