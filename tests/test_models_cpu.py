"""Model-zoo smoke tests (CPU, small batch): every zoo model builds,
runs one train_iter + update, validates, and checkpoints round-trip."""

import numpy as np
import pytest
import torch

ZOO = [
    ("theanompi_amd.models.alex_net", "AlexNet", 2),
    ("theanompi_amd.models.googlenet", "GoogLeNet", 2),
    ("theanompi_amd.models.resnet50", "ResNet50", 2),
    ("theanompi_amd.models.vgg16", "VGG16", 2),
    ("theanompi_amd.models.wresnet", "WResNet", 4),
    ("theanompi_amd.models.cifar10", "Cifar10_model", 8),
    ("theanompi_amd.models.test_model", "TestModel", 8),
]


def make(modname, clsname, batch, device="cpu"):
    import importlib
    cls = getattr(importlib.import_module(modname), clsname)
    m = cls({"rank": 0, "size": 1, "device": device, "verbose": False,
             "batch_size": batch, "n_iters": 2, "val_iters": 1})
    m.compile_iter_fns("grad")
    return m


@pytest.mark.parametrize("modname,clsname,batch", ZOO)
def test_model_one_step_cpu(modname, clsname, batch):
    torch.manual_seed(0)
    m = make(modname, clsname, batch)
    loss, err = m.train_iter(0)
    m.apply_update()
    assert np.isfinite(float(loss))
    assert 0.0 <= float(err) <= 1.0
    c, t1, t5 = m.val_iter(0)
    assert np.isfinite(c)


def test_checkpoint_roundtrip_bn_model(tmp_path):
    """WRN has BatchNorm params — checkpoint must cover gamma/beta too."""
    torch.manual_seed(1)
    m = make("theanompi_amd.models.wresnet", "WResNet", 4)
    m.train_iter(0)
    m.apply_update()
    m.save(str(tmp_path), epoch=0)
    ref = [p.detach().clone() for p in m.params]
    torch.manual_seed(2)
    m2 = make("theanompi_amd.models.wresnet", "WResNet", 4)
    m2.load(str(tmp_path), epoch=0)
    for a, b in zip(ref, m2.params):
        assert torch.allclose(a, b, atol=1e-6)


@pytest.mark.gpu
@pytest.mark.parametrize("modname,clsname,batch", ZOO)
def test_model_trains_gpu(modname, clsname, batch):
    torch.manual_seed(0)
    m = make(modname, clsname, max(batch, 4), device="cuda:0")
    l0, _ = m.train_iter(0)
    m.apply_update()
    l1, _ = m.train_iter(1)
    m.apply_update()
    assert np.isfinite(float(l0)) and np.isfinite(float(l1))


@pytest.mark.gpu
def test_imagenet_loader_gpu(tmp_path):
    """Shard loader end-to-end on GPU (pinned ring -> u8_finish)."""
    from theanompi_amd.models.data.imagenet import (ImageNetData,
                                                    write_synthetic_shards)
    paths = write_synthetic_shards(str(tmp_path), n_shards=3,
                                   images_per_shard=8, hw=40)
    d = ImageNetData(paths, paths[:1], batch_size=8, crop=32,
                     n_classes=1000, device="cuda:0")
    try:
        x, y = d.get_train_batch(0)
        assert x.shape == (8, 32, 32, 4)
        assert x.dtype == torch.bfloat16
        assert float(x[..., 3].abs().max()) == 0.0      # pad channel
        assert y.is_cuda
    finally:
        d.close()


def test_checkpoint_cross_rule(tmp_path):
    """A checkpoint written while training under one rule loads under
    another (same per-param .npy format)."""
    torch.manual_seed(3)
    m = make("theanompi_amd.models.test_model", "TestModel", 8)
    m.train_iter(0)
    m.apply_update()
    m.save(str(tmp_path), epoch=2)
    ref = [p.detach().clone() for p in m.params]
    m2 = make("theanompi_amd.models.test_model", "TestModel", 8)
    m2.compile_iter_fns("avg")           # the EASGD workers' mode
    m2.load(str(tmp_path), epoch=2)
    for a, b in zip(ref, m2.params):
        assert torch.allclose(a, b, atol=1e-6)


@pytest.mark.gpu
def test_convergence_gpu():
    """End-to-end learning check: on a LEARNABLE synthetic task (class
    templates + noise) the bf16 GPU pipeline must reach ~0 train error
    — exercises every kernel's numerics jointly, not just per-op
    tolerances."""
    import torch as T
    from theanompi_amd.models.cifar10 import Cifar10_model
    T.manual_seed(0)
    m = make("theanompi_amd.models.cifar10", "Cifar10_model", 128,
             device="cuda:0")
    d = m.data
    templates = T.randint(0, 256, (10, 32, 32, 4), dtype=T.int16)
    templates[..., 3:] = 0
    with T.no_grad():
        y_all = d.y.cpu()
        noise = T.randint(-60, 60, d.x.shape, dtype=T.int16)
        newx = (templates[y_all] + noise).clamp(0, 255).to(T.uint8)
        newx[..., 3:] = 0
        d.x.copy_(newx.to(d.x.device))
    err = 1.0
    for i in range(200):
        loss, err = m.train_iter(i % d.n_batch_train)
        m.apply_update()
    assert float(err) < 0.05, f"did not learn: err={float(err)}"


def test_zoo_alias_paths():
    """Reference-layout import paths keep working."""
    from theanompi_amd.models.lasagne_model_zoo.vgg import VGG16
    from theanompi_amd.models.lasagne_model_zoo.resnet50 import ResNet50
    from theanompi_amd.models.keras_model_zoo.wresnet import WResNet
    assert VGG16.name == "vgg16"
    assert ResNet50.name == "resnet50"
    assert WResNet.name == "wresnet"


@pytest.mark.gpu
def test_alexnet_shard_pipeline_gpu(tmp_path):
    """AlexNet trains from the real shard pipeline on GPU: loader
    child -> pinned ring -> u8_finish -> HIP conv stack."""
    from theanompi_amd.models.data.imagenet import write_synthetic_shards
    paths = write_synthetic_shards(str(tmp_path), n_shards=3,
                                   images_per_shard=8, hw=232)
    import importlib
    cls = getattr(importlib.import_module("theanompi_amd.models.alex_net"),
                  "AlexNet")
    m = cls({"rank": 0, "size": 1, "device": "cuda:0", "verbose": False,
             "batch_size": 8, "train_shards": ",".join(paths),
             "val_shards": paths[0]})
    try:
        m.compile_iter_fns("grad")
        l0, _ = m.train_iter(0)
        m.apply_update()
        l1, _ = m.train_iter(1)
        m.apply_update()
        assert np.isfinite(float(l0)) and np.isfinite(float(l1))
        c, t1, t5 = m.val_iter(0)
        assert np.isfinite(c)
    finally:
        m.data.close()


def test_copy_slice_no_engine_copies():
    """The concat-write inception path must not trigger autograd's
    CopySlices materialization (profiles/glue_attribution_r2.txt:
    mark_dirty on a VIEW cost two engine copies per module).
    copy_slice routes through a non-view alias; backward must produce
    correct grads with ZERO aten.copy_ calls."""
    import collections
    from torch.utils._python_dispatch import TorchDispatchMode
    from theanompi_amd.models import layers as L

    class Count(TorchDispatchMode):
        def __init__(self):
            super().__init__()
            self.c = collections.Counter()

        def __torch_dispatch__(self, func, types, args=(), kwargs=None):
            self.c[str(func)] += 1
            return func(*args, **(kwargs or {}))

    x1 = torch.randn(2, 4, 4, 8, requires_grad=True)
    x2 = torch.randn(2, 4, 4, 8, requires_grad=True)
    buf = torch.empty(2, 4, 4, 16)
    y1 = L.copy_slice(buf[..., :8], x1 * 2)
    y2 = L.copy_slice(buf[..., 8:], x2 * 3)
    # autograd re-wraps a returned input as a view OF THE ALIAS —
    # what matters is that neither output is a view of the shared
    # concat buffer (that would rebase buf onto CopySlices)
    assert y1._base is not buf and y2._base is not buf
    out = L.join_views(buf, y1, y2)
    m = Count()
    with m:
        out.sum().backward()
    copies = sum(v for k, v in m.c.items() if "copy_" in k)
    assert copies == 0, dict(m.c)
    assert torch.allclose(x1.grad, torch.full_like(x1, 2.0))
    assert torch.allclose(x2.grad, torch.full_like(x2, 3.0))
    # and the written buffer holds the branch outputs
    assert torch.equal(out[..., :8], (x1 * 2).detach())
    assert torch.equal(out[..., 8:], (x2 * 3).detach())
