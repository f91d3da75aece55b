"""Optimizer builders — rebuild of ``theanompi/lib/opt.py`` [pub]
(SURVEY.md §2.3): SGD + momentum (+ weight decay), with the cdd
split-phase formulation around the BSP comm point:

  phase 1 (pre-exchange):  vel = mu*vel - lr*(grad + wd*w)
  exchange:                vel <- mean over workers (BSP_Exchanger)
  phase 2 (post-exchange): w  += vel

'avg'/'grad' modes use the fused single step.  On GPU the update is one
pass of the fused HIP kernel N12 over each flat fp32 master buffer,
optionally writing the bf16 working copy in the same pass.
"""

import torch

from theanompi_amd import ops as _ops


class ParamRec:
    """One trainable: fp32 master + vel (+ optional bf16 mirror)."""

    __slots__ = ("master", "vel", "bf16_out", "name", "lr_mult", "wd_mult")

    def __init__(self, master, name="", lr_mult=1.0, wd_mult=1.0,
                 bf16_out=None):
        self.master = master          # fp32 tensor with .grad after bwd
        self.vel = torch.zeros_like(master.data)
        self.bf16_out = bf16_out      # bf16 tensor refreshed by the step
        self.name = name
        self.lr_mult = lr_mult
        self.wd_mult = wd_mult


class SGDMomentum:
    """Momentum SGD over ParamRecs, usable under every rule."""

    def __init__(self, recs, lr=0.01, momentum=0.9, weight_decay=0.0,
                 grad_arena=None):
        self.recs = list(recs)
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        # flat pre-zeroed buffer the conv/FC backward kernels atomically
        # accumulate into (one fill per step instead of one per op)
        self.grad_arena = grad_arena
        self._gpu = any(r.master.is_cuda for r in self.recs)
        self.persistent_grads = False   # graph mode: stable grad buffers
        self._desc = None               # cached kernel descriptor

    def make_persistent(self):
        """Pre-allocate .grad buffers with stable addresses and build the
        fused-update descriptor once (required for hipGraph capture; also
        saves the per-step descriptor upload in eager mode).  Changing lr
        afterwards requires rebuild (adjust_hyperp calls invalidate())."""
        self.persistent_grads = True
        for r in self.recs:
            if r.master.grad is None:
                r.master.grad = torch.zeros_like(r.master.data)
        self._desc = None

    def invalidate(self):
        self._desc = None

    # -- helpers --------------------------------------------------------------

    def zero_grad(self):
        if self.persistent_grads:
            for r in self.recs:
                r.master.grad.zero_()
            return
        if self.grad_arena is not None:
            self.grad_arena.zero_()      # ONE fill for every conv/FC grad
        for r in self.recs:
            if r.master.grad is not None:
                r.master.grad = None

    def grads(self):
        out = []
        for r in self.recs:
            g = r.master.grad
            if g is None:
                g = torch.zeros_like(r.master.data)
                r.master.grad = g
            out.append(g)
        return out

    def vels(self):
        return [r.vel for r in self.recs]

    def _kernel_step(self, r, grad, phase):
        C = _ops.ext()
        out = r.bf16_out if (r.bf16_out is not None and phase != 1) \
            else torch.empty(0, dtype=torch.bfloat16,
                             device=r.master.device)
        C.sgd_step(r.master.data.view(-1), grad.view(-1), r.vel.view(-1),
                   out.view(-1),
                   self.lr * r.lr_mult, self.momentum,
                   self.weight_decay * r.wd_mult, int(phase))

    def _torch_step(self, r, grad, phase):
        lr = self.lr * r.lr_mult
        wd = self.weight_decay * r.wd_mult
        with torch.no_grad():
            if phase in (0, 1):
                g = grad
                if wd:
                    g = g.add(r.master.data, alpha=wd)
                r.vel.mul_(self.momentum).add_(g, alpha=-lr)
            if phase == 0:
                r.master.data.add_(r.vel)
            elif phase == 2:
                r.master.data.add_(grad)   # grad holds the averaged vels
            if r.bf16_out is not None and phase != 1:
                r.bf16_out.copy_(r.master.data)

    def _dispatch(self, r, grad, phase):
        if r.master.is_cuda:
            self._kernel_step(r, grad, phase)
        else:
            self._torch_step(r, grad, phase)

    # -- public API ------------------------------------------

    def _batched_gpu(self, phase, grad_of):
        """One kernel launch for every GPU rec (N12 batched)."""
        C = _ops.ext()
        ws, gs, vs, outs, lrs, wds = [], [], [], [], [], []
        self._empty_bf16 = getattr(self, "_empty_bf16", None)
        for r in self.recs:
            if not r.master.is_cuda:
                g = grad_of(r)
                if g is not None:
                    self._torch_step(r, g, phase)
                continue
            g = grad_of(r)
            if g is None:
                continue
            if self._empty_bf16 is None:
                self._empty_bf16 = torch.empty(
                    0, dtype=torch.bfloat16, device=r.master.device)
            ws.append(r.master.data.view(-1))
            gs.append(g.view(-1))
            vs.append(r.vel.view(-1))
            outs.append(r.bf16_out.view(-1)
                        if (r.bf16_out is not None and phase != 1)
                        else self._empty_bf16)
            lrs.append(self.lr * r.lr_mult)
            wds.append(self.weight_decay * r.wd_mult)
        if ws:
            C.sgd_step_multi(ws, gs, vs, outs, lrs, wds,
                             self.momentum, phase)

    def _cached_step(self, phase):
        C = _ops.ext()
        if self._desc is None:
            ws, gs, vs, outs, lrs, wds = [], [], [], [], [], []
            empty = torch.empty(0, dtype=torch.bfloat16,
                                device=self.recs[0].master.device)
            for r in self.recs:
                ws.append(r.master.data.view(-1))
                gs.append(r.master.grad.view(-1))
                vs.append(r.vel.view(-1))
                outs.append(r.bf16_out.view(-1)
                            if r.bf16_out is not None else empty)
                lrs.append(self.lr * r.lr_mult)
                wds.append(self.weight_decay * r.wd_mult)
            self._desc = C.sgd_multi_make(ws, gs, vs, outs, lrs, wds)
            self._nseg = len(ws)
            self._maxn4 = max(w.numel() // 4 for w in ws)
        C.sgd_multi_run(self._desc, self._nseg, self._maxn4,
                        self.momentum, phase)

    def step(self):
        """Full fused step: v = mu*v - lr*(g + wd*w); w += v."""
        if self._gpu and self.persistent_grads:
            self._cached_step(0)
        elif self._gpu:
            self._batched_gpu(0, lambda r: r.master.grad)
        else:
            for r in self.recs:
                if r.master.grad is not None:
                    self._dispatch(r, r.master.grad, 0)

    def phase1(self):
        """cdd pre-exchange: compute velocities only."""
        if self._gpu:
            self._batched_gpu(1, lambda r: r.master.grad)
        else:
            for r in self.recs:
                if r.master.grad is not None:
                    self._dispatch(r, r.master.grad, 1)

    def phase2(self):
        """cdd post-exchange: apply the (exchanged, averaged) velocities."""
        if self._gpu:
            self._batched_gpu(2, lambda r: r.vel)
        else:
            for r in self.recs:
                self._dispatch(r, r.vel, 2)
