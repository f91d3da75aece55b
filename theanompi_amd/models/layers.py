"""Layer library — rebuild of Theano-MPI's ``theanompi/models/layers2.py``
[pub] as NHWC modules over the HIP op set.

GPU path (bf16 compute, fp32 master):
  * Conv2d / Linear keep their master weights as fp32 tensors in the
    kernels' flat padded layout ([G, Kgp, Kdf] — see ops/hip/conv.hip),
    refresh a bf16 working copy plus a repacked dgrad copy after every
    optimizer step, and run the implicit-GEMM HIP kernels through a
    torch.autograd.Function so fp32 grads land in ``weight.grad``.
  * Channel dims are physically padded (ops.functional.cpad_for); padded
    weight entries are zero and stay zero (their activations are zero or
    their gradients are zero by construction).

CPU path (fp32, plumbing/test mode): same modules dispatch to torch
reference ops (ops/functional.py *_ref).
"""

import math
import os

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from theanompi_amd import ops as _ops
from theanompi_amd.ops import functional as OF
from theanompi_amd.ops.functional import cpad_for

# ---------------------------------------------------------------------------
# gather-table / repack-map builders (host side, run once per layer)
# ---------------------------------------------------------------------------

# dual-output relu backward (masked dy + pre-padded dgrad halo in one
# pass) — THEANOMPI_DUAL_RELU=0 disables for A/B measurement
DUAL_RELU = os.environ.get("THEANOMPI_DUAL_RELU", "1") != "0"

PIECE = 16   # flat-k granularity in elements (32 B of bf16)
KPAD = 64    # flat reduction dim padded to one GEMM K-tile


def _build_spans(R, S, Cg, Cphys, combine_sc):
    """Return (spans, flat_slots, elem2flat) where spans is a list of
    (dh, dw, c0, length) contiguous runs in the input tensor, covering
    every (r, s, c) reduction element; flat_slots is the padded flat
    K-dim; elem2flat maps (r, s, c) -> flat index."""
    spans = []
    elem2flat = {}
    base = 0
    if combine_sc:
        # one run per r covering all (s, c): contiguous when Cg == Cphys
        for r in range(R):
            length = S * Cg
            spans.append((r, 0, 0, length, base))
            for s in range(S):
                for c in range(Cg):
                    elem2flat[(r, s, c)] = base + s * Cg + c
            base += (length + PIECE - 1) // PIECE * PIECE
        base = (base + KPAD - 1) // KPAD * KPAD
        return spans, base, elem2flat
    if True:
        for r in range(R):
            for s in range(S):
                length = Cg
                spans.append((r, s, 0, length, base))
                for c in range(Cg):
                    elem2flat[(r, s, c)] = base + c
                base += (length + PIECE - 1) // PIECE * PIECE
    base = (base + KPAD - 1) // KPAD * KPAD
    return spans, base, elem2flat


def _spans_to_table(spans, flat_slots, Wp, Cphys):
    """int32 element-offset per 16-piece from the (group-adjusted) pixel
    base pointer.  Padding pieces point at the span start (their weights
    are zero, so the gathered values are irrelevant)."""
    n_pieces = flat_slots // PIECE
    table = np.zeros(n_pieces, dtype=np.int32)
    for (dh, dw, c0, length, base) in spans:
        run_start = dh * Wp * Cphys + dw * Cphys + c0
        n_slots = (length + PIECE - 1) // PIECE * PIECE
        for j in range(n_slots // PIECE):
            pos = j * PIECE
            if pos < length:
                table[(base // PIECE) + j] = run_start + pos
            else:
                table[(base // PIECE) + j] = run_start
    return table


class ConvPlan:
    """All layout/dispatch state for one conv (or linear) layer on GPU.

    Forward GEMM:   y[M=N*P*Q, Kgp]  = gatherA(xp)[M, Kdf] @ Wflat[Kdf, Kgp]
    Dgrad GEMM:     dx[M=N*H*W, Cgp] = gatherA(dyp)[M, Kdf2] @ Wdg[Kdf2, Cgp]
                    (per (sh x sw) output class when stride > 1)
    Wgrad GEMM(TN): dW[Kdf, Kgp]    += gatherA(xp)[Mpix, Kdf]^T @ dy[Mpix, Kgp]
    """

    def __init__(self, N, H, W, C, K, R, S, stride, pad, groups, device):
        sh, sw = stride
        ph, pw = pad
        self.NHWC = (N, H, W, C)
        self.R, self.S, self.G = R, S, groups
        self.stride, self.pad = stride, pad
        self.Cg = C // groups
        self.Cphys = C
        assert C % groups == 0
        assert self.Cg == cpad_for(self.Cg) or groups == 1, \
            "per-group channels must be physically padded"
        self.Kg = K // groups            # logical K per group
        self.Kgp = cpad_for(self.Kg)
        self.Kphys = groups * self.Kgp
        self.P = (H + 2 * ph - R) // sh + 1
        self.Q = (W + 2 * pw - S) // sw + 1
        self.Hp, self.Wp = H + 2 * ph, W + 2 * pw
        self.device = device

        combine = (groups == 1)
        spans, self.Kdf, self.e2f = _build_spans(R, S, self.Cg, C, combine)
        tab = _spans_to_table(spans, self.Kdf, self.Wp, C)
        self.table = torch.from_numpy(tab).to(device)
        self.kdf_mask = self.build_kdf_mask() if str(device) != "cpu" \
            else None

        # ---- dgrad plans: one per (sh x sw) output class --------------------
        self.dgrad_plans = []
        Hd = self.P + 0  # dy logical dims
        for i in range(sh):
            for j in range(sw):
                Hi = (H - 1 - i) // sh + 1 if i < H else 0
                Wj = (W - 1 - j) // sw + 1 if j < W else 0
                if Hi <= 0 or Wj <= 0:
                    continue
                r_list = [r for r in range(R) if (i + ph - r) % sh == 0]
                s_list = [s for s in range(S) if (j + pw - s) % sw == 0]
                dr = {r: (i + ph - r) // sh for r in r_list}
                ds = {s: (j + pw - s) // sw for s in s_list}
                # halo needed on dy so all (p' + dr) are in range
                if not r_list or not s_list:
                    # this class gets no contribution (possible when
                    # kernel doesn't cover the class) -> zero dx there
                    self.dgrad_plans.append(None)
                    continue
                halo_t = max(0, -min(dr.values()))
                halo_b = max(0, (Hi - 1) + max(dr.values()) - (self.P - 1))
                halo_l = max(0, -min(ds.values()))
                halo_r = max(0, (Wj - 1) + max(ds.values()) - (self.Q - 1))
                Qp = self.Q + halo_l + halo_r
                spans2 = []
                base = 0
                e2f2 = {}
                for r in r_list:
                    for s in s_list:
                        spans2.append((dr[r] + halo_t, ds[s] + halo_l, 0,
                                       self.Kgp, base))
                        for k in range(self.Kgp):
                            e2f2[(r, s, k)] = base + k
                        base += self.Kgp
                base = (base + KPAD - 1) // KPAD * KPAD
                tab2 = _spans_to_table(spans2, base, Qp, self.Kphys)
                self.dgrad_plans.append({
                    "cls": (i, j), "Hi": Hi, "Wj": Wj,
                    "halo": (halo_t, halo_b, halo_l, halo_r),
                    "Kdf2": base, "e2f2": e2f2,
                    "table": torch.from_numpy(tab2).to(device),
                })
        # every input pixel's class has a contributing plan -> the dgrad
        # launches write ALL of dx and the buffer needn't be pre-zeroed
        self.dgrad_full_cover = all(p is not None
                                    for p in self.dgrad_plans)
        # repack maps (dest flat [G,Kdf2,Cgp] <- src flat [G,Kdf,Kgp])
        self._repack_maps = None

    # -- weight layout ------------------------------------------------------

    def flat_shape(self):
        # out-channel-major so the GEMM's B tile is k-contiguous per row
        return (self.G, self.Kgp, self.Kdf)

    def w_to_flat(self, w_kcrs, out=None):
        """logical [K, Cg, R, S] fp32 -> flat [G, Kgp, Kdf]."""
        G, Kg = self.G, self.Kg
        flat = out if out is not None else torch.zeros(
            self.flat_shape(), dtype=torch.float32, device=w_kcrs.device)
        flat.zero_()
        for r in range(self.R):
            for s in range(self.S):
                fd0 = self.e2f[(r, s, 0)]
                for g in range(G):
                    blk = w_kcrs[g * Kg:(g + 1) * Kg, :, r, s]  # [Kg, Cg]
                    flat[g, :Kg, fd0:fd0 + self.Cg] = blk
        return flat

    def flat_to_w(self, flat):
        G, Kg = self.G, self.Kg
        w = torch.zeros((G * Kg, self.Cg, self.R, self.S),
                        dtype=torch.float32, device=flat.device)
        for r in range(self.R):
            for s in range(self.S):
                fd0 = self.e2f[(r, s, 0)]
                for g in range(G):
                    w[g * Kg:(g + 1) * Kg, :, r, s] = \
                        flat[g, :Kg, fd0:fd0 + self.Cg]
        return w

    def build_kdf_mask(self):
        """f32 [Kdf]: 1 at real (r,s,c) slots, 0 at padding slots — used
        to zero wgrad's contributions at padded flat positions so padded
        master weights stay exactly zero."""
        m = torch.zeros(self.Kdf, dtype=torch.float32)
        for fd in self.e2f.values():
            m[fd] = 1.0
        return m.to(self.device)

    def build_repack_maps(self):
        """int32 maps dest-linear -> src-linear (-1 => zero) for each
        dgrad class; used by _C.gather_bf16 after every optimizer step."""
        if self._repack_maps is not None:
            return self._repack_maps
        # FC / 1x1 s1 conv: the repack is a plain transpose of the flat
        # weight — use the LDS transpose kernel instead of an int32 map
        # (the map gather was 5% of the AlexNet step and its numpy
        # construction dominates model build time for big FCs).
        live = [p for p in self.dgrad_plans if p is not None]
        if (self.R == 1 and self.S == 1 and self.G == 1
                and len(live) == 1 and len(self.dgrad_plans) == 1):
            self._repack_maps = ["transpose"]
            return self._repack_maps
        maps = []
        G, Kgp, Kg = self.G, self.Kgp, self.Kg
        src_strideG = Kgp * self.Kdf
        for plan in self.dgrad_plans:
            if plan is None:
                maps.append(None)
                continue
            Kdf2 = plan["Kdf2"]
            m = np.full((G, self.Cg, Kdf2), -1, dtype=np.int32)
            car = np.arange(self.Cg, dtype=np.int32)
            for (r, s, k), fd2 in plan["e2f2"].items():
                if k >= Kg:
                    continue  # padded k: stays zero
                fd = self.e2f[(r, s, 0)]
                # dest[g, c, fd2] = w_flat[g, k, fd + c]
                for g in range(G):
                    m[g, :, fd2] = (g * src_strideG + k * self.Kdf
                                    + fd + car)
            mt = torch.from_numpy(m).to(self.device)
            maps.append(mt)
        self._repack_maps = maps
        return maps


# ---------------------------------------------------------------------------
# GPU autograd Function for conv / linear
# ---------------------------------------------------------------------------

class _EmptyF32Cache:
    def __init__(self):
        self._t = {}

    def get(self, device):
        t = self._t.get(device)
        if t is None:
            t = torch.empty(0, dtype=torch.float32, device=device)
            self._t[device] = t
        return t


_EMPTY_F32 = _EmptyF32Cache()


def _arena_alias(v):
    """Non-view tensor aliasing the same storage as view ``v``."""
    t = torch.empty(0, dtype=v.dtype, device=v.device)
    t.set_(v.untyped_storage(), v.storage_offset(), v.shape)
    return t


class _JoinViews(torch.autograd.Function):
    """Declare that ``buf`` is composed of the given channel slices
    (already written in place by concat-writing convs).  Forward is a
    no-op; backward splits dy into channel-slice VIEWS, which the
    branch convs' relu backward consumes directly — the whole
    concat/split round-trip costs zero copies."""

    @staticmethod
    def forward(ctx, buf, *slices):
        ctx.widths = [sl.shape[-1] for sl in slices]
        return buf

    @staticmethod
    def backward(ctx, dy):
        outs = []
        c0 = 0
        for w in ctx.widths:
            outs.append(dy[..., c0:c0 + w])
            c0 += w
        return (None, *outs)


def join_views(buf, *slices):
    return _JoinViews.apply(buf, *slices)


def _row_strided(t):
    """Last-dim-contiguous view whose rows sit a uniform stride apart
    (e.g. a channel slice of a concat buffer)."""
    if t.dim() != 4 or t.stride(3) != 1:
        return False
    ld = t.stride(2)
    return (ld >= t.shape[3] and t.stride(1) == ld * t.shape[2]
            and t.stride(0) == t.stride(1) * t.shape[1])


class _HipConvFn(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, w_master, b_master, layer):
        C = _ops.ext()
        plan = layer.plan
        ph, pw = plan.pad
        xp = C.halo_pad(x, ph, pw, 0.0) if (ph or pw) else x.contiguous()
        N = x.shape[0]
        M = N * plan.P * plan.Q
        stats = None
        if getattr(layer, "_collect_stats", False) and layer.training:
            if layer._stats_buf is None:
                layer._stats_buf = torch.zeros(
                    32 * 2 * layer.plan.Kphys, dtype=torch.float32,
                    device=x.device)
            stats = layer._stats_buf
        out = getattr(layer, "_out_slice", None)
        if out is not None:
            # concat-writing conv: the epilogue writes a channel slice
            # of the caller's concat buffer directly (uniform row
            # stride) — torch.cat disappears.  Restricted to fused-relu
            # convs: the backward's strided-y support lives in the relu
            # mask kernel.
            layer._out_slice = None
            assert layer.relu and out.shape == (N, plan.P, plan.Q,
                                                plan.Kphys)
            y = out
        else:
            y = torch.empty((N, plan.P, plan.Q, plan.Kphys),
                            dtype=torch.bfloat16, device=x.device)
        vf = layer._v_fwd
        if vf is None or layer._v_M != M:
            from theanompi_amd.ops import tuning
            vf = tuning.variant_for("fwd", M, plan.Kgp, plan.Kdf, plan.G)
            layer._v_fwd = vf
            layer._v_M = M
            layer._v_wgrad = tuning.wgrad_for(M, plan.Kgp, plan.Kdf,
                                              plan.G)
        layer._stats_valid = C.conv_fwd(
            xp, layer.w_bf16,
            layer.b_f32 if layer.b_f32 is not None else
            _EMPTY_F32.get(x.device),
            plan.table, y,
            plan.Wp * plan.Cphys,                # pixel row stride (h)
            plan.Cphys,                          # pixel col stride (w)
            plan.stride[0], plan.stride[1],
            plan.Cg if plan.G > 1 else 0,        # group in-chan offset
            bool(layer.relu), stats, vf)
        ctx.save_for_backward(xp, y if layer.relu else torch.empty(0))
        ctx.layer = layer
        return y

    @staticmethod
    def backward(ctx, dy):
        C = _ops.ext()
        layer = ctx.layer
        plan = layer.plan
        xp, y = ctx.saved_tensors
        layer._dyp_ready = None
        if layer.relu:
            # the relu mask kernel accepts a channel slice of a concat
            # buffer directly (uniform row stride), so the inception
            # backward's split needs no .contiguous() copy
            if not (dy.is_contiguous() or _row_strided(dy)):
                dy = dy.contiguous()
            dym = torch.empty(dy.shape, dtype=dy.dtype,
                              device=dy.device)
            single = layer._dgrad_single_plan() if DUAL_RELU else None
            if (single is not None and ctx.needs_input_grad[0]
                    and dy.shape[-1] % 8 == 0):
                # dual write: dym (wgrad/bias) + the halo-padded copy
                # dgrad needs, skipping its halo_pad pass+launch
                ht, hb, hl, hr = single["halo"]
                N, P, Q, K = dym.shape
                buf = getattr(layer, "_dyp_buf", None)
                if buf is None or buf.shape[0] != N:
                    buf = torch.zeros((N, P + ht + hb, Q + hl + hr, K),
                                      dtype=torch.bfloat16,
                                      device=dy.device)
                    layer._dyp_buf = buf
                C.relu_bwd_dual(dy, y, dym, buf, ht, hl)
                layer._dyp_ready = buf
            else:
                C.relu_bwd(dy, y, dym)
            dy = dym
        else:
            dy = dy.contiguous()
        # bias grad.  The arena slices (_dw_buf/_db_buf) are pre-zeroed
        # by zero_grad; they are only safe to hand to autograd while
        # .grad is None (otherwise AccumulateGrad would add the slice
        # into itself), so manual grad accumulation falls back to a
        # fresh zeroed buffer.  _arena_alias re-wraps the slice as a
        # fresh NON-view alias: AccumulateGrad deep-copies any returned
        # grad that is a view (one copyBuffer per parameter per step),
        # but steals a single-ref non-view tensor.
        db = None
        if layer.b_f32 is not None:
            db = getattr(layer, "_db_buf", None)
            db = _arena_alias(db) if (db is not None
                                      and layer.bias.grad is None) else \
                torch.zeros_like(layer.b_f32)
        # weight grad (fp32, flat layout); bias grad rides the same
        # kernel (fTile==0 blocks reduce their staged dy tiles)
        dw = getattr(layer, "_dw_buf", None)
        dw = _arena_alias(dw) if (dw is not None
                                  and layer.weight.grad is None) else \
            torch.zeros(plan.flat_shape(), dtype=torch.float32,
                        device=dy.device)
        wv, wt = getattr(layer, "_v_wgrad", (0, 0)) or (0, 0)
        C.conv_wgrad(xp, dy, plan.table, dw, plan.kdf_mask,
                     plan.Wp * plan.Cphys, plan.Cphys,
                     plan.stride[0], plan.stride[1],
                     plan.Cg if plan.G > 1 else 0, wv, wt, db)
        dx = None
        if ctx.needs_input_grad[0]:
            dx = layer._dgrad(dy)
        return dx, dw, db, None


# ---------------------------------------------------------------------------
# modules
# ---------------------------------------------------------------------------

class _Base(nn.Module):
    """Common bits: device/gpu flag + checkpoint hooks."""

    # tuned kernel variants (ops.tuning), cached per observed GEMM-M —
    # shared by every layer that routes through _HipConvFn (Conv2d AND
    # Linear-as-1x1-conv)
    _v_fwd = None
    _v_M = -1
    _v_wgrad = (0, 0)

    def logical_params(self):
        """[(name, getter()->np-compatible tensor, setter(tensor))] in the
        reference's per-layer W-then-b order."""
        return []

    def _dgrad_single_plan(self):
        """The sole dgrad plan when the layer has exactly one stride
        class WITH a halo (the pre-padded dual-relu-backward target);
        None otherwise.  Lives on _Base: Linear shares the conv
        backward path."""
        cached = getattr(self, "_dgrad_single_c", -1)
        if cached != -1:
            return cached
        plans = [p2 for p2 in getattr(self, "plan").dgrad_plans
                 if p2 is not None]
        single = None
        if len(plans) == 1 and any(plans[0]["halo"]):
            single = plans[0]
        self._dgrad_single_c = single
        return single

    def post_step(self):
        """Called after every optimizer step (refresh bf16 copies)."""


def _init_tensor(shape, init, fan_in=None, fan_out=None):
    t = torch.empty(*shape, dtype=torch.float32)
    kind, *arg = (init if isinstance(init, tuple) else (init,))
    if kind == "normal":
        t.normal_(0.0, arg[0] if arg else 0.01)
    elif kind == "he":
        t.normal_(0.0, math.sqrt(2.0 / fan_in))
    elif kind == "xavier":
        a = math.sqrt(6.0 / (fan_in + fan_out))
        t.uniform_(-a, a)
    elif kind == "constant":
        t.fill_(arg[0] if arg else 0.0)
    else:
        raise ValueError(kind)
    return t


class Conv2d(_Base):
    """NHWC conv (+ optional fused ReLU).

    GPU: implicit-GEMM HIP kernels (ops/hip/conv.hip; reference N1-N3 in
    SURVEY.md §2.6).  CPU: torch reference.  Weights logical [K, Cg, R, S].
    """

    def __init__(self, in_shape, K, R, S=None, stride=1, pad=0, groups=1,
                 relu=False, init=("he",), bias_init=0.0, device="cpu",
                 batch=None, needs_dgrad=True, has_bias=True,
                 collect_stats=False):
        super().__init__()
        # collect_stats: the epilogue also produces bucketed per-channel
        # sum/sum-sq of the output for a following BatchNorm (fused
        # stats; consumer fetches via pop_stats())
        self._collect_stats = collect_stats
        self._stats_buf = None
        self._stats_valid = False
        self.needs_dgrad = needs_dgrad
        self.has_bias = has_bias
        S = S if S is not None else R
        stride = (stride, stride) if isinstance(stride, int) else stride
        pad = (pad, pad) if isinstance(pad, int) else pad
        H, W, Cin = in_shape            # logical input HWC (phys C on GPU)
        self.K, self.R, self.S = K, R, S
        self.stride, self.pad, self.groups = stride, pad, groups
        self.relu = relu
        self.device_t = torch.device(device)
        self.gpu = self.device_t.type == "cuda"
        self.Cg_logical = Cin // groups

        fan_in = self.Cg_logical * R * S
        w0 = _init_tensor((K, self.Cg_logical, R, S), init, fan_in,
                          K * R * S // groups)
        b0 = torch.full((K,), float(bias_init))

        if self.gpu:
            Cphys = groups * cpad_for(self.Cg_logical)
            self.Cphys_in = Cphys
            self.plan = ConvPlan(batch or 1, H, W, Cphys, K, R, S,
                                 stride, pad, groups, self.device_t)
            wl = torch.zeros((K, Cphys // groups, R, S))
            wl[:, :self.Cg_logical] = w0
            flat = self.plan.w_to_flat(wl.to(self.device_t))
            self.weight = nn.Parameter(flat)
            bp = torch.zeros(self.plan.Kphys, device=self.device_t)
            if has_bias:
                for g in range(groups):
                    bp[g * self.plan.Kgp:
                       g * self.plan.Kgp
                       + self.plan.Kg] = b0[g * self.plan.Kg:
                                            (g + 1) * self.plan.Kg]
            self.bias = nn.Parameter(bp) if has_bias else None
            self.register_buffer("w_bf16",
                                 flat.to(torch.bfloat16).contiguous())
            self.b_f32 = self.bias.data if has_bias else None
            if not needs_dgrad:      # first layer: dx never computed
                self.plan.dgrad_plans = []
                self.plan._repack_maps = []
            maps = self.plan.build_repack_maps()
            self._wdg = self._alloc_wdg(maps)
            self.out_shape = (self.plan.P, self.plan.Q, self.plan.Kphys)
            self.out_logical_c = K
            self.post_step()          # fill w_bf16 + dgrad repack copies
        else:
            self.weight = nn.Parameter(w0)
            self.bias = nn.Parameter(b0) if has_bias else None
            P = (H + 2 * pad[0] - R) // stride[0] + 1
            Q = (W + 2 * pad[1] - S) // stride[1] + 1
            self.out_shape = (P, Q, K)
            self.out_logical_c = K

    def forward(self, x, out=None):
        if self.gpu:
            if out is not None:
                self._out_slice = out
            return _HipConvFn.apply(x, self.weight, self.bias, self)
        c_log = self.Cg_logical * self.groups
        if x.shape[-1] != c_log:
            x = x[..., :c_log]
        return OF.conv2d_ref(x, self.weight, self.bias, self.stride,
                             self.pad, self.groups, self.relu)

    _collect_stats = False          # class defaults: Linear shares the
    _stats_buf = None               # conv path but not __init__
    _stats_valid = False

    def pop_stats(self):
        """Bucketed output statistics from the last forward (fused BN),
        or None if the epilogue could not collect them (split-K path /
        alternate tiles) — the consumer then falls back to bn_sums."""
        if not self._stats_valid:
            return None
        self._stats_valid = False
        return self._stats_buf

    def _dgrad(self, dy):
        C = _ops.ext()
        plan = self.plan
        N = dy.shape[0]
        H, W = plan.NHWC[1], plan.NHWC[2]
        alloc = torch.empty if plan.dgrad_full_cover else torch.zeros
        dx = alloc((N, H, W, plan.Cphys), dtype=torch.bfloat16,
                   device=dy.device)
        pre = getattr(self, "_dyp_ready", None)
        self._dyp_ready = None
        for plan2, wdg in zip(plan.dgrad_plans, self._wdg):
            if plan2 is None:
                continue
            ht, hb, hl, hr = plan2["halo"]
            if pre is not None:
                dyp = pre                 # dual relu backward pre-padded
            else:
                dyp = C.halo_pad_lrtb(dy, ht, hb, hl, hr, 0.0) \
                    if (ht or hb or hl or hr) else dy
            i, j = plan2["cls"]
            Qp = plan.Q + hl + hr
            vd = plan2.get("variant")
            if vd is None or plan2.get("vN") != N:
                from theanompi_amd.ops import tuning
                vd = tuning.variant_for(
                    "dgrad", N * plan2["Hi"] * plan2["Wj"], plan.Cg,
                    plan2["Kdf2"], plan.G)
                plan2["variant"] = vd
                plan2["vN"] = N
            C.conv_dgrad(dyp, wdg, plan2["table"], dx,
                         Qp * plan.Kphys, plan.Kphys,
                         plan2["Hi"], plan2["Wj"],
                         i, j, plan.stride[0], plan.stride[1],
                         plan.Kgp if plan.G > 1 else 0,
                         plan.Cg, vd)
        return dx

    def _alloc_wdg(self, maps):
        out = []
        for i, m in enumerate(maps):
            if m is None:
                out.append(None)
            elif isinstance(m, str):          # "transpose"
                plan2 = self.plan.dgrad_plans[i]
                out.append(torch.zeros(
                    (self.plan.G, self.plan.Cg, plan2["Kdf2"]),
                    dtype=torch.bfloat16, device=self.device_t))
            else:
                out.append(torch.zeros(m.shape, dtype=torch.bfloat16,
                                       device=self.device_t))
        return out

    def repack_dgrad(self, C):
        """Refresh the dgrad weight copies from w_bf16."""
        for m, wdg in zip(self.plan.build_repack_maps(), self._wdg):
            if m is None:
                continue
            if isinstance(m, str):            # transpose path
                plan = self.plan
                C.transpose_bf16(self.w_bf16.view(-1), wdg.view(-1),
                                 plan.Kgp, plan.Cg, plan.Kdf,
                                 wdg.shape[-1])
            else:
                C.gather_bf16(self.w_bf16.view(-1), m.view(-1),
                              wdg.view(-1))

    def post_step(self):
        if not self.gpu:
            return
        C = _ops.ext()
        self.w_bf16.copy_(self.weight.data)   # fp32 -> bf16 cast copy
        if self.bias is not None:
            self.b_f32 = self.bias.data
        self.repack_dgrad(C)

    def logical_params(self):
        if self.gpu:
            out = [("W", lambda: self.plan.flat_to_w(self.weight.data)
                                     [:, :self.Cg_logical],
                    self._set_w)]
            if self.bias is not None:
                out.append(("b", lambda: self._get_b(), self._set_b))
            return out
        out = [("W", lambda: self.weight.data,
                lambda v: self.weight.data.copy_(v))]
        if self.bias is not None:
            out.append(("b", lambda: self.bias.data,
                        lambda v: self.bias.data.copy_(v)))
        return out

    def _get_b(self):
        plan = self.plan
        out = torch.empty(self.K, device=self.device_t)
        for g in range(self.groups):
            out[g * plan.Kg:(g + 1) * plan.Kg] = \
                self.bias.data[g * plan.Kgp:g * plan.Kgp + plan.Kg]
        return out

    def _set_b(self, v):
        plan = self.plan
        for g in range(self.groups):
            self.bias.data[g * plan.Kgp:g * plan.Kgp + plan.Kg] = \
                v[g * plan.Kg:(g + 1) * plan.Kg]
        self.post_step()

    def _set_w(self, v):
        plan = self.plan
        wl = torch.zeros((self.K, plan.Cg, self.R, self.S))
        wl[:, :self.Cg_logical] = v.cpu()
        plan.w_to_flat(wl.to(self.device_t), out=self.weight.data)
        self.post_step()


class MultiConv1x1(Conv2d):
    """Several 1x1 convs over the SAME input fused into one GEMM
    (inception branch batching, GPU only): output channels are the
    branches' padded blocks ``[pad(k0) | pad(k1) | ...]``; callers
    slice per branch.  One fwd GEMM, one dgrad (which also fuses the
    branches' dx sum), one wgrad, one bias pass — GoogLeNet-32b is
    launch-latency bound (615 dispatches/step measured), so fewer,
    fatter launches are the lever multi-stream could not be.

    Numerics match the unmerged branches exactly: each branch's pad
    rows start at zero (like an unmerged conv's Kg..Kgp rows) and train
    under the same gradients.  ``branch_logical_params()`` exposes each
    branch's logical ``[k_i, Cin]``-shaped weight/bias so checkpoints
    keep the reference's per-branch entries (weights are cross-loadable
    with an unmerged model; momentum files remain layout-positional)."""

    def __init__(self, in_shape, ks, **kw):
        from theanompi_amd.ops.functional import cpad_for
        self.branch_k = list(ks)
        self.branch_kp = [cpad_for(k) for k in ks]
        self.branch_off = [sum(self.branch_kp[:i])
                           for i in range(len(ks))]
        super().__init__(in_shape, sum(self.branch_kp), 1, **kw)
        assert self.gpu and self.groups == 1
        assert self.plan.Kg == self.plan.Kgp == sum(self.branch_kp)
        # branch pad rows start at zero, exactly like an unmerged
        # conv's physical pad rows
        with torch.no_grad():
            for k, kp, off in zip(self.branch_k, self.branch_kp,
                                  self.branch_off):
                if kp > k:
                    self.weight.data[0, off + k:off + kp].zero_()
                    self.bias.data[off + k:off + kp].zero_()
        self.post_step()

    def slices(self, y):
        return [y[..., o:o + kp]
                for o, kp in zip(self.branch_off, self.branch_kp)]

    def branch_logical_params(self):
        """Per-branch [(W, get, set), (b, get, set)] lists in branch
        order; shapes match the equivalent standalone Conv2d."""
        outs = []
        for i, (k, off) in enumerate(zip(self.branch_k,
                                         self.branch_off)):
            def get_w(k=k, off=off):
                full = self.plan.flat_to_w(self.weight.data)
                return full[off:off + k, :self.Cg_logical]

            def set_w(v, k=k, off=off):
                plan = self.plan
                wl = self.plan.flat_to_w(self.weight.data).cpu()
                wl[off:off + k, :self.Cg_logical] = v.cpu()
                plan.w_to_flat(wl.to(self.device_t),
                               out=self.weight.data)
                self.post_step()

            def get_b(k=k, off=off):
                return self.bias.data[off:off + k].clone()

            def set_b(v, k=k, off=off):
                self.bias.data[off:off + k] = v.to(self.device_t)
                self.post_step()

            outs.append([(f"W{i}", get_w, set_w),
                         (f"b{i}", get_b, set_b)])
        return outs

    def logical_params(self):
        # flattened branch order (used when the owner does not compose
        # a custom order)
        out = []
        for entry in self.branch_logical_params():
            out.extend(entry)
        return out


class _CopySlice(torch.autograd.Function):
    """Write src into a channel slice of a (non-tracked) concat buffer
    and return the slice as an autograd output: gradient passes
    straight through to src."""

    @staticmethod
    def forward(ctx, dst_slice, src):
        with torch.no_grad():
            dst_slice.copy_(src)
        return dst_slice

    @staticmethod
    def backward(ctx, dy):
        return None, dy


def _alias_strided(v):
    """Non-view tensor over the same storage/strides as view ``v``.
    mark_dirty on a VIEW rebases the base's graph onto CopySlices,
    which materializes TWO engine copies per inception module in
    backward (the b1 slice grad + the whole concat-buffer grad,
    ~101 MB/step on GoogLeNet — profiles/glue_attribution_r2.txt);
    a non-view alias keeps autograd's view machinery out entirely."""
    t = torch.empty(0, dtype=v.dtype, device=v.device)
    t.set_(v.untyped_storage(), v.storage_offset(), v.shape, v.stride())
    return t


def copy_slice(dst_slice, src):
    if dst_slice._base is not None:
        dst_slice = _alias_strided(dst_slice)
    return _CopySlice.apply(dst_slice, src)


class _SplitChannels(torch.autograd.Function):
    """Channel split whose backward is ONE cat: plain slicing would
    give each consumer a SliceBackward (zeros_like + scatter) plus adds
    to combine them — ~8 full-size kernels per inception module."""

    @staticmethod
    def forward(ctx, m, *widths):
        ctx.widths = widths
        ctx.shape = m.shape
        ctx.dev = m.device
        ctx.dt = m.dtype
        outs, o = [], 0
        for w in widths:
            outs.append(m[..., o:o + w])
            o += w
        assert o == m.shape[-1], "widths must partition the channels"
        return tuple(outs)

    @staticmethod
    def backward(ctx, *ds):
        parts = []
        for d, w in zip(ds, ctx.widths):
            if d is None:
                d = torch.zeros(*ctx.shape[:-1], w, dtype=ctx.dt,
                                device=ctx.dev)
            parts.append(d)
        return (torch.cat(parts, dim=-1), *([None] * len(ctx.widths)))


def split_channels(m, widths):
    return _SplitChannels.apply(m, *widths)


class Linear(_Base):
    """FC layer; logical weight [IN, OUT] (Theano layout).  GPU path is a
    1x1x1 image conv through the same implicit-GEMM kernel (N11)."""

    def __init__(self, n_in, n_out, relu=False, init=("he",),
                 bias_init=0.0, device="cpu", batch=None):
        super().__init__()
        self.n_in, self.n_out = n_in, n_out
        self.relu = relu
        self.device_t = torch.device(device)
        self.gpu = self.device_t.type == "cuda"
        w0 = _init_tensor((n_in, n_out), init, n_in, n_out)
        b0 = torch.full((n_out,), float(bias_init))
        if self.gpu:
            inp = cpad_for(n_in)
            self.plan = ConvPlan(batch or 1, 1, 1, inp, n_out, 1, 1,
                                 (1, 1), (0, 0), 1, self.device_t)
            flat = torch.zeros(self.plan.flat_shape(), device=self.device_t)
            flat[0, :n_out, :n_in] = w0.t().to(self.device_t)
            self.weight = nn.Parameter(flat)
            bp = torch.zeros(self.plan.Kphys, device=self.device_t)
            bp[:n_out] = b0
            self.bias = nn.Parameter(bp)
            self.register_buffer("w_bf16",
                                 flat.to(torch.bfloat16).contiguous())
            self.b_f32 = self.bias.data
            maps = self.plan.build_repack_maps()
            self._wdg = self._alloc_wdg(maps)
            self.in_phys = inp
            self.out_features_phys = self.plan.Kphys
            self.post_step()          # fill w_bf16 + dgrad repack copies
        else:
            self.weight = nn.Parameter(w0)
            self.bias = nn.Parameter(b0)
            self.out_features_phys = n_out

    def forward(self, x):
        if self.gpu:
            n = x.shape[0]
            if x.shape[1] != self.in_phys:
                x = F.pad(x, (0, self.in_phys - x.shape[1]))
            x4 = x.reshape(n, 1, 1, self.in_phys)
            y = _HipConvFn.apply(x4, self.weight, self.bias, self)
            return y.view(n, -1)
        if x.shape[-1] != self.n_in:
            x = x[..., :self.n_in]
        return OF.linear_ref(x, self.weight, self.bias, self.relu)

    _dgrad = Conv2d._dgrad
    post_step = Conv2d.post_step
    _alloc_wdg = Conv2d._alloc_wdg
    repack_dgrad = Conv2d.repack_dgrad

    def logical_params(self):
        if self.gpu:
            return [("W", lambda: self.weight.data[0, :self.n_out,
                                                   :self.n_in].t(),
                     self._set_w),
                    ("b", lambda: self.bias.data[:self.n_out],
                     self._set_b)]
        return [("W", lambda: self.weight.data,
                 lambda v: self.weight.data.copy_(v)),
                ("b", lambda: self.bias.data,
                 lambda v: self.bias.data.copy_(v))]

    def _set_w(self, v):
        self.weight.data.zero_()
        self.weight.data[0, :self.n_out, :self.n_in] = \
            v.t().to(self.device_t)
        self.post_step()

    def _set_b(self, v):
        self.bias.data.zero_()
        self.bias.data[:self.n_out] = v.to(self.device_t)
        self.post_step()


class MaxPool2d(_Base):
    def __init__(self, kernel, stride, pad=0):
        super().__init__()
        self.kernel, self.stride, self.pad = kernel, stride, pad

    def forward(self, x):
        return OF.maxpool2d(x, self.kernel, self.stride, self.pad)


class AvgPool2d(_Base):
    def __init__(self, kernel, stride, pad=0):
        super().__init__()
        self.kernel, self.stride, self.pad = kernel, stride, pad

    def forward(self, x):
        return OF.avgpool2d(x, self.kernel, self.stride, self.pad)


class LRN(_Base):
    """Cross-map local response normalisation (AlexNet/GoogLeNet stem)."""

    def __init__(self, size=5, alpha=1e-4, beta=0.75, k=1.0):
        super().__init__()
        self.size, self.alpha, self.beta, self.k = size, alpha, beta, k

    def forward(self, x):
        return OF.lrn(x, self.size, self.alpha, self.beta, self.k)


class BatchNorm2d(_Base):
    def __init__(self, c_phys, eps=1e-5, momentum=0.1, relu=False,
                 device="cpu"):
        super().__init__()
        dev = torch.device(device)
        self.gamma = nn.Parameter(torch.ones(c_phys, device=dev))
        self.beta = nn.Parameter(torch.zeros(c_phys, device=dev))
        self.register_buffer("running_mean",
                             torch.zeros(c_phys, device=dev))
        self.register_buffer("running_var",
                             torch.ones(c_phys, device=dev))
        self.eps, self.momentum, self.relu = eps, momentum, relu

    def forward(self, x, prestats=None, residual=None, relu=None):
        return OF.batchnorm(x, self.gamma, self.beta, self.running_mean,
                            self.running_var, self.training, self.momentum,
                            self.eps,
                            self.relu if relu is None else relu,
                            prestats=prestats, residual=residual)

    def logical_params(self):
        return [("gamma", lambda: self.gamma.data,
                 lambda v: self.gamma.data.copy_(v)),
                ("beta", lambda: self.beta.data,
                 lambda v: self.beta.data.copy_(v))]


class Dropout(_Base):
    def __init__(self, p=0.5):
        super().__init__()
        self.p = p

    def forward(self, x):
        return OF.dropout(x, self.p, self.training)


class ReLU(_Base):
    def forward(self, x):
        return OF.relu(x)


class Flatten(_Base):
    def forward(self, x):
        return x.reshape(x.shape[0], -1)
