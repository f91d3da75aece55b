"""Contract adapter for out-of-tree models.

The reference proved its model contract is framework-agnostic by
wrapping a *Keras* model (``models/keras_model_zoo/wresnet.py`` [pub],
SURVEY.md §2.4): any object exposing the duck-type contract trains under
any rule.  This module is that capability for the MI355X rebuild: wrap a
plain ``torch.nn.Module`` — written with stock ``torch.nn`` layers, no
BaseModel inheritance, no knowledge of this framework — and train it
under BSP / EASGD / GOSGD unchanged.

    from theanompi_amd.models.module_adapter import wrap_module

    class MyNet(torch.nn.Module): ...          # any foreign model code

    MyModel = wrap_module(lambda: MyNet(), batch_size=32, lr=0.01,
                          image_hw=32, n_classes=10)
    # session script:
    rule.init(devices=[...], modelfile='my_session_module',
              modelclass='MyModel')

The adapter deliberately does NOT route through the framework's HIP
layer library: the wrapped module runs under whatever kernels it was
written with (stock PyTorch ops here), while the *parallelism* — the
exchangers, rules, recorder, checkpoint format — is the framework's.
That separation is exactly what the reference's Keras wrap demonstrated.
"""

import torch
import torch.nn.functional as F

from theanompi_amd.lib import helper_funcs
from theanompi_amd.lib.opt import ParamRec, SGDMomentum
from theanompi_amd.models.data.synthetic import SyntheticData


class ContractAdapter:
    """Implements the model contract (SURVEY.md §1 L4->L2) around a
    plain nn.Module: params/vels/recs/data + compile_iter_fns,
    train_iter, val_iter, apply_update, adjust_hyperp, save/load,
    reset_iter, refresh_all."""

    #: hyperparameter defaults; overridden per wrapped class
    batch_size = 32
    n_epochs = 2
    lr = 0.01
    momentum = 0.9
    weight_decay = 0.0
    n_classes = 10
    image_hw = 32
    channels_last = False     # True: feed [B,H,W,C] instead of [B,C,H,W]

    def module_factory(self):
        raise NotImplementedError

    def data_factory(self):
        """Default data: deterministic synthetic batches shaped for the
        wrapped module.  Override for real data."""
        return SyntheticData(
            batch_size=self.batch_size, image_hw=self.image_hw,
            n_classes=self.n_classes, device=self.device, c_phys=3,
            n_train_batches=int(self.config.get("n_iters", 8)),
            n_val_batches=int(self.config.get("val_iters", 2)),
            rank=self.rank, size=self.size, use_loader=False)

    def __init__(self, config):
        self.config = dict(config)
        self.rank = int(config.get("rank", 0))
        self.size = int(config.get("size", 1))
        self.device = torch.device(config.get("device", "cpu"))
        self.gpu = self.device.type == "cuda"
        self.verbose = bool(config.get("verbose", self.rank == 0))
        if "batch_size" in config:
            self.batch_size = int(config["batch_size"])
        if "epochs" in config:
            self.n_epochs = int(config["epochs"])
        self.current_lr = self.lr
        self.epoch = 0
        self.sync_type = "cdd"

        self.module = self.module_factory().to(self.device)
        self.module.train()
        self.data = self.data_factory()
        self.recs = [ParamRec(p, name)
                     for name, p in self.module.named_parameters()]
        self.params = [r.master for r in self.recs]
        self.vels = [r.vel for r in self.recs]
        self.opt = None

    # -- contract: compile -----------------------------------------------------

    def compile_iter_fns(self, sync_type="cdd"):
        self.sync_type = sync_type
        self.opt = SGDMomentum(self.recs, lr=self.current_lr,
                               momentum=self.momentum,
                               weight_decay=self.weight_decay)

    # -- contract: train / validate ---------------------------------------------

    def _input(self, x):
        # framework data objects serve NHWC; stock torch modules want NCHW
        if not self.channels_last and x.dim() == 4 and x.shape[-1] <= 4:
            x = x.permute(0, 3, 1, 2).contiguous()
        return x

    def train_iter(self, count, recorder=None):
        x, y = self.data.get_train_batch(count)
        if recorder:
            recorder.end("wait")
        self.opt.zero_grad()
        logits = self.module(self._input(x))
        loss = F.cross_entropy(logits[:, :self.n_classes], y)
        loss.backward()
        with torch.no_grad():
            err = (logits[:, :self.n_classes].argmax(1) != y).float().mean()
        if self.sync_type == "cdd":
            self.opt.phase1()
        elif self.sync_type == "avg":
            self.opt.step()
        if recorder:
            recorder.end("calc")
            recorder.train_error(count, float(loss), float(err))
        return float(loss), float(err)

    def apply_update(self):
        if self.sync_type == "cdd":
            self.opt.phase2()
        elif self.sync_type == "grad":
            self.opt.step()

    def val_iter(self, count, recorder=None):
        self.module.eval()
        try:
            with torch.no_grad():
                x, y = self.data.get_val_batch(count)
                logits = self.module(self._input(x))[:, :self.n_classes]
                loss = F.cross_entropy(logits, y)
                top5 = logits.topk(min(5, self.n_classes), dim=1).indices
                top1 = (top5[:, 0] != y).float().mean()
                t5 = (~(top5 == y[:, None]).any(dim=1)).float().mean()
        finally:
            self.module.train()
        return float(loss), float(top1), float(t5)

    # -- contract: schedule / checkpoint / glue ---------------------------------

    def lr_steps(self):
        return (20, 40, 60)

    def adjust_hyperp(self, epoch):
        self.epoch = epoch
        decays = sum(1 for e in self.lr_steps() if epoch >= e)
        self.current_lr = self.lr * (0.1 ** decays)
        if self.opt:
            self.opt.lr = self.current_lr

    def save(self, directory, epoch=None):
        epoch = self.epoch if epoch is None else epoch
        helper_funcs.save_weights(self.params, directory, epoch)
        helper_funcs.save_momentums(self.vels, directory, epoch)

    def load(self, directory, epoch=None):
        epoch = self.epoch if epoch is None else epoch
        helper_funcs.load_weights(self.params, directory, epoch)
        helper_funcs.load_momentums(self.vels, directory, epoch)

    def reset_iter(self, mode):
        if hasattr(self.data, "reset"):
            self.data.reset(mode)

    def refresh_all(self):
        """No derived weight copies: the wrapped module trains directly
        on the fp32 masters."""


def wrap_module(factory, **hyperparams):
    """Build a contract class around a plain nn.Module factory.

    The returned class is instantiable with the worker's config dict and
    importable as a ``modelclass`` by every rule."""
    def module_factory(self):
        return factory()

    attrs = dict(hyperparams)
    attrs["module_factory"] = module_factory
    return type("WrappedModule", (ContractAdapter,), attrs)


# ---------------------------------------------------------------------------
# Demo / test subject: a CNN written entirely with stock torch.nn — the
# rebuild's analog of the reference's keras_model_zoo Wide-ResNet.
# ---------------------------------------------------------------------------

def _plain_cnn():
    import torch.nn as nn
    return nn.Sequential(
        nn.Conv2d(3, 16, 3, padding=1), nn.ReLU(),
        nn.MaxPool2d(2),
        nn.Conv2d(16, 32, 3, padding=1), nn.ReLU(),
        nn.AdaptiveAvgPool2d(4),
        nn.Flatten(),
        nn.Linear(32 * 16, 10),
    )


PlainTorchCNN = wrap_module(_plain_cnn, batch_size=8, n_epochs=1,
                            lr=0.05, n_classes=10, image_hw=16)
PlainTorchCNN.__name__ = "PlainTorchCNN"
