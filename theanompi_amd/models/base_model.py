"""Generic implementation of the model contract (SURVEY.md §1 L4->L2).

Concrete zoo models (alex_net.py, googlenet.py, ...) subclass this and
provide ``build_module()`` + hyperparameters; anything else satisfying
the duck-type contract also works (the workers only use the contract).
"""

import os
from contextlib import contextmanager as _contextmanager

import torch

from theanompi_amd.lib import helper_funcs
from theanompi_amd.lib.opt import ParamRec, SGDMomentum
from theanompi_amd.models import layers as L
from theanompi_amd.ops import functional as OF


_ROCTX = os.environ.get("THEANOMPI_AMD_ROCTX", "0") not in ("0", "")


class _rtx:
    """roctx range around an iteration phase — visible in
    rocprofv3 --marker-trace; torch.cuda.nvtx maps to roctx on ROCm.
    Enabled with THEANOMPI_AMD_ROCTX=1, zero-cost otherwise."""

    def __init__(self, name):
        self.name = name

    def __enter__(self):
        if _ROCTX:
            torch.cuda.nvtx.range_push(self.name)

    def __exit__(self, *exc):
        if _ROCTX:
            torch.cuda.nvtx.range_pop()
        return False


class BaseModel:
    """Implements: params, vels, data, compile_iter_fns, train_iter,
    val_iter, adjust_hyperp, load, save, reset_iter."""

    # subclass knobs
    name = "base"
    batch_size = 128
    n_epochs = 60
    lr = 0.01
    momentum = 0.9
    weight_decay = 5e-4
    n_classes = 1000

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

        self.module = self.build_module()       # returns an nn.Module
        self.module.train()
        self.data = self.build_data()
        self._build_recs()
        self.opt = None

    # -- to be provided by subclasses -----------------------------------------

    def build_module(self):
        raise NotImplementedError

    def build_data(self):
        raise NotImplementedError

    def forward(self, x):
        """Return logits [B, n_classes(+pad)]."""
        return self.module(x)

    # -- parameter bookkeeping -------------------------------------------------

    def _build_recs(self):
        self.recs = []
        self._post_step_layers = []
        for name, mod in self.module.named_modules():
            if isinstance(mod, (L.Conv2d, L.Linear)):
                bf16 = getattr(mod, "w_bf16", None)
                self.recs.append(ParamRec(mod.weight, name + ".W",
                                          bf16_out=bf16))
                if mod.bias is not None:
                    self.recs.append(ParamRec(mod.bias, name + ".b",
                                              wd_mult=0.0))
                if mod.gpu:
                    self._post_step_layers.append(mod)
            elif isinstance(mod, L.BatchNorm2d):
                self.recs.append(ParamRec(mod.gamma, name + ".gamma",
                                          wd_mult=0.0))
                self.recs.append(ParamRec(mod.beta, name + ".beta",
                                          wd_mult=0.0))
        # velocities live in one flat arena so the BSP cdd exchange is a
        # single zero-copy all-reduce (no per-tensor flatten copies)
        total = sum(r.master.numel() for r in self.recs)
        self.vel_arena = torch.zeros(total, dtype=torch.float32,
                                     device=self.device)
        off = 0
        for r in self.recs:
            n = r.master.numel()
            r.vel = self.vel_arena[off:off + n].view(r.master.shape)
            off += n
        # pre-zeroed grad arena: the GPU conv/FC backward accumulates
        # atomically and would otherwise pay one fill launch per op
        # (~14% of the GoogLeNet step); instead zero_grad clears this
        # arena once and backward hands out slices.
        self.grad_arena = None
        if self.gpu:
            self.grad_arena = torch.zeros(total, dtype=torch.float32,
                                          device=self.device)
            off = 0
            views = {}
            for r in self.recs:
                n = r.master.numel()
                views[id(r.master)] = \
                    self.grad_arena[off:off + n].view(r.master.shape)
                off += n
            for mod in self.module.modules():
                if isinstance(mod, (L.Conv2d, L.Linear)):
                    mod._dw_buf = views.get(id(mod.weight))
                    if mod.bias is not None:
                        mod._db_buf = views.get(id(mod.bias))
        # reference-parity flat views for the exchangers / checkpoints
        self.params = [r.master for r in self.recs]
        self.vels = [r.vel for r in self.recs]

    def _refresh_layers(self):
        """Repack dgrad weight copies after a weight change (GPU).
        All layers' repacks run as ONE batched gather launch (the
        per-layer launches were 5.8% of the AlexNet step)."""
        from theanompi_amd import ops as _ops
        if not self._post_step_layers:
            return
        C = _ops.ext()
        if not hasattr(self, "_repack_desc"):
            srcs, maps, dsts = [], [], []
            self._transpose_layers = []
            for mod in self._post_step_layers:
                for m, wdg in zip(mod.plan.build_repack_maps(), mod._wdg):
                    if m is None:
                        continue
                    if isinstance(m, str):
                        self._transpose_layers.append((mod, wdg))
                    else:
                        srcs.append(mod.w_bf16.view(-1))
                        maps.append(m.view(-1))
                        dsts.append(wdg.view(-1))
            if srcs:
                self._repack_desc = C.gather_multi_make(srcs, maps, dsts)
                self._repack_nseg = len(srcs)
                self._repack_maxn = max(d.numel() for d in dsts)
            else:
                self._repack_desc = None
            # transpose-path layers batch into ONE launch too (they
            # were one launch per 1x1 conv per step — 46/step on
            # GoogLeNet, pure launch latency)
            self._trans_desc = None
            if self._transpose_layers:
                tsrcs, tdsts, rows, cols, sst, dst_ = [], [], [], [], [], []
                for mod, wdg in self._transpose_layers:
                    plan = mod.plan
                    tsrcs.append(mod.w_bf16.view(-1))
                    tdsts.append(wdg.view(-1))
                    rows.append(plan.Kgp)
                    cols.append(plan.Cg)
                    sst.append(plan.Kdf)
                    dst_.append(wdg.shape[-1])
                import torch as _t
                self._trans_buf = _t.empty(
                    len(tsrcs) * 48, dtype=_t.uint8,
                    device=tsrcs[0].device)
                mr, mc = C.transpose_multi_make(
                    tsrcs, tdsts, rows, cols, sst, dst_, self._trans_buf)
                self._trans_desc = (len(tsrcs), mr, mc)
        if self._repack_desc is not None:
            C.gather_multi_run(self._repack_desc, self._repack_nseg,
                               self._repack_maxn)
        if self._trans_desc is not None:
            n, mr, mc = self._trans_desc
            C.transpose_multi_run(self._trans_buf, n, mr, mc)

    def refresh_all(self):
        """Resync every derived weight copy after an out-of-band change to
        the fp32 masters (exchange, merge, broadcast, checkpoint load)."""
        with torch.no_grad():
            for rec in self.recs:
                if rec.bf16_out is not None:
                    rec.bf16_out.copy_(rec.master.data)
        self._refresh_layers()

    # -- contract: compile ----------------------------------------------------

    def compile_iter_fns(self, sync_type="cdd"):
        """Reference compiled Theano functions here (minutes); we build the
        optimizer and warm the kernels (instant)."""
        self.sync_type = sync_type
        self.opt = SGDMomentum(self.recs, lr=self.current_lr,
                               grad_arena=self.grad_arena,
                               momentum=self.momentum,
                               weight_decay=self.weight_decay)

    def sync_tensors(self):
        """What the BSP exchanger moves, by sync mode."""
        if self.sync_type == "cdd":
            return self.vels
        if self.sync_type == "avg":
            return self.params
        if self.sync_type == "grad":
            return self.opt.grads()
        raise ValueError(self.sync_type)

    # -- contract: train ------------------------------------------------------

    def train_iter(self, count, recorder=None):
        with _rtx("load"):
            x, y = self.data.get_train_batch(count)
        if recorder:
            recorder.end("wait")
        with _rtx("fwd_bwd"):
            loss, err = self._fwd_bwd(x, y)
        if self.sync_type == "cdd":
            self.opt.phase1()
        elif self.sync_type == "avg":
            self.opt.step()
        if recorder:
            recorder.end("calc")
            recorder.train_error(count, loss, err)
        return loss, err

    def _fwd_bwd(self, x, y):
        self.opt.zero_grad()
        logits = self.forward(x)
        # loss mean + top-1 err ride the xent kernel (fused reductions)
        loss, _probs, err = OF.softmax_xent_err(logits, y, self.n_classes)
        loss.backward()
        return loss.detach(), err

    def apply_update(self):
        """Called by the worker after the exchange (phase 2 of cdd; no-op
        weight math for avg; full fused step for grad mode)."""
        ctx = _rtx("update")
        ctx.__enter__()
        try:
            return self._apply_update_inner()
        finally:
            ctx.__exit__()

    def _apply_update_inner(self):
        if self.sync_type == "cdd":
            self.opt.phase2()
        elif self.sync_type == "grad":
            self.opt.step()
        self._refresh_layers()

    # -- contract: validate ---------------------------------------------------

    def val_iter(self, count, recorder=None):
        self.module.eval()
        try:
            with torch.no_grad():
                x, y = self.data.get_val_batch(count)
                logits = self.forward(x)[:, :self.n_classes].float()
                loss, _ = OF.softmax_xent_ref(logits, y, self.n_classes)
                top5 = logits.topk(5, dim=1).indices
                top1err = (top5[:, 0] != y).float().mean()
                top5err = (~(top5 == y[:, None]).any(dim=1)).float().mean()
        finally:
            self.module.train()
        if recorder:
            recorder.end("calc")
        return float(loss), float(top1err), float(top5err)

    # -- contract: schedule / checkpoint --------------------------------------

    def adjust_hyperp(self, epoch):
        """Step LR schedule (reference models stepped /10)."""
        self.epoch = epoch
        decays = sum(1 for e in self.lr_steps() if epoch >= e)
        self.current_lr = self.lr * (0.1 ** decays)
        if self.opt:
            self.opt.lr = self.current_lr
            if getattr(self.opt, "invalidate", None):
                self.opt.invalidate()

    def lr_steps(self):
        return (20, 40, 60)

    def _logical_param_list(self):
        """Checkpoint entries in reference order.  A module with
        ``owns_logical_params`` composes its whole subtree's entries
        itself (e.g. the inception block keeps the unmerged per-branch
        order even when its 1x1s run as one fused GEMM)."""
        out = []
        owned = []
        for name, mod in self.module.named_modules():
            if any(name == p or name.startswith(p + ".")
                   for p in owned):
                continue
            if getattr(mod, "owns_logical_params", False):
                owned.append(name)
                out.extend(mod.logical_params())
            elif isinstance(mod, L._Base):
                out.extend(mod.logical_params())
        return out

    @_contextmanager
    def _vels_as_params(self):
        """Temporarily swap every rec's master .data with its velocity
        buffer so the LOGICAL getters/setters address the momenta: the
        V_<i>_<epoch>.npy files get the same per-parameter shapes and
        reference order as the W files (SURVEY §5.4 — true SGD-state
        resume in the reference's logical layout), including through
        merged layers.  Callers must refresh derived copies after a
        swap that wrote (load does)."""
        saved = []
        try:
            for r in self.recs:
                saved.append(r.master.data)
                r.master.data = r.vel
            yield
        finally:
            for r, d in zip(self.recs, saved):
                r.master.data = d

    def save(self, directory, epoch=None):
        epoch = self.epoch if epoch is None else epoch
        plist = self._logical_param_list()
        logical = [g() for (_n, g, _s) in plist]
        helper_funcs.save_weights(logical, directory, epoch)
        with self._vels_as_params():
            vlogical = [g() for (_n, g, _s) in plist]
        helper_funcs.save_weights(vlogical, directory, epoch,
                                  prefix="V")

    def load(self, directory, epoch=None):
        epoch = self.epoch if epoch is None else epoch
        plist = self._logical_param_list()
        import numpy as np
        import os
        for i, (_n, _g, setter) in enumerate(plist):
            arr = np.load(os.path.join(directory, f"W_{i}_{epoch}.npy"))
            setter(torch.from_numpy(arr))
        # refresh bf16/dgrad copies from the restored weights BEFORE the
        # vel swap (the setters' post_step inside the swap refreshes
        # from the velocities; redo it after)
        with self._vels_as_params():
            for i, (_n, _g, setter) in enumerate(plist):
                arr = np.load(
                    os.path.join(directory, f"V_{i}_{epoch}.npy"))
                setter(torch.from_numpy(arr))
        self.refresh_all()

    def imagenet_or_synthetic(self, crop, synthetic_factory):
        """Real shard pipeline when the session passed
        ``train_shards``/``val_shards`` (comma-separated .npy shard
        paths, reference workflow SURVEY §2.5); synthetic otherwise."""
        shards = self.config.get("train_shards", "")
        if not shards:
            return synthetic_factory()
        from theanompi_amd.models.data.imagenet import ImageNetData
        train = [p for p in str(shards).split(",") if p]
        val = [p for p in
               str(self.config.get("val_shards", "")).split(",") if p] \
            or train[:1]
        mean = self.config.get("mean_path", "") or None
        return ImageNetData(train, val, batch_size=self.batch_size,
                            crop=crop, n_classes=self.n_classes,
                            device=self.device, rank=self.rank,
                            size=self.size,
                            c_phys=4 if self.gpu else 3,
                            mean_path=mean)

    def reset_iter(self, mode):
        if hasattr(self.data, "reset"):
            self.data.reset(mode)
