"""GPU kernel numerics: every HIP kernel vs a plain PyTorch fp32
reference on the same bf16-quantized inputs (SURVEY.md §4 rebuild test
strategy).  All tests need an MI355X -> @pytest.mark.gpu."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def dev():
    return torch.device("cuda", 0)


def q(x):
    """quantize to bf16 grid, keep fp32 (for references)."""
    return x.to(torch.bfloat16).float()


def rel_err(got, ref):
    got = got.detach().float().cpu()
    ref = ref.detach().float().cpu()
    denom = ref.abs().max().clamp_min(1e-6)
    return ((got - ref).abs().max() / denom).item()


# ---------------------------------------------------------------------------
# probe + primitives
# ---------------------------------------------------------------------------

def test_mfma_layout_probe():
    from theanompi_amd import ops as O
    lay = O.ext().mfma_layout()
    assert lay in (0, 1)


def test_relu_add_pack():
    from theanompi_amd import ops as O
    C = O.ext()
    x = torch.randn(1000, 16, device=dev()).to(torch.bfloat16)
    y = torch.empty_like(x)
    C.relu_fwd(x, y)
    assert torch.equal(y.float(), x.float().clamp_min(0))

    a = torch.randn(64, 32, device=dev()).to(torch.bfloat16)
    b = torch.randn(64, 32, device=dev()).to(torch.bfloat16)
    out = torch.empty_like(a)
    C.add_relu(a, b, out, True)
    ref = (a.float() + b.float()).clamp_min(0)
    assert rel_err(out, ref) < 2e-2

    src = torch.randn(4096, device=dev())
    dst = torch.empty(4096, dtype=torch.bfloat16, device=dev())
    C.pack_bf16(src, dst)
    assert torch.equal(dst, src.to(torch.bfloat16))
    back = torch.empty(4096, device=dev())
    C.unpack_bf16(dst, back)
    assert torch.equal(back, dst.float())


def test_segsum():
    """N14 segment-sum (asa_xgmi phase 2) vs torch fp32 reference."""
    from theanompi_amd import ops as O
    C = O.ext()
    for w in (2, 7, 8):
        seg = 4 * 1000
        x = torch.randn(w * seg, device=dev())
        out = torch.empty(seg, device=dev())
        C.segsum(x, out, w)
        ref = x.view(w, seg).sum(0)
        assert torch.allclose(out, ref, atol=1e-5), (w, seg)
        xb = x.to(torch.bfloat16)
        ob = torch.empty(seg, dtype=torch.bfloat16, device=dev())
        C.segsum(xb, ob, w)
        # fp32 accumulation: must match the fp32 sum of the bf16 inputs
        refb = xb.float().view(w, seg).sum(0).to(torch.bfloat16)
        assert torch.equal(ob, refb), w


def test_halo_pad():
    from theanompi_amd import ops as O
    C = O.ext()
    x = torch.randn(2, 5, 7, 8, device=dev()).to(torch.bfloat16)
    y = C.halo_pad(x, 2, 1, 0.0)
    assert y.shape == (2, 9, 9, 8)
    assert torch.equal(y[:, 2:7, 1:8].float(), x.float())
    assert y[:, :2].float().abs().sum() == 0
    assert y[:, 7:].float().abs().sum() == 0
    y2 = C.halo_pad_lrtb(x, 1, 2, 3, 0, float("-inf"))
    assert y2.shape == (2, 8, 10, 8)
    assert torch.isinf(y2[:, 0].float()).all()


# ---------------------------------------------------------------------------
# conv / linear via the layer library (fwd + full backward)
# ---------------------------------------------------------------------------

def conv_case(B, H, W, Cin, K, R, stride, pad, groups=1, relu=False,
              atol=0.02):
    # Tolerance tier: every compared path here is
    # bf16 inputs with FP32 ACCUMULATION against a reference computed
    # from identically bf16-quantized operands, so the only divergence
    # is accumulation order + one bf16 output rounding (~4e-3 ulp);
    # 2e-2 on a max-normalized error leaves ~4x headroom (margins
    # printed below, checked on hardware) while still catching any
    # systematic bias the old 6e-2 would have hidden.
    from theanompi_amd.models import layers as L
    from theanompi_amd.ops import functional as OF
    torch.manual_seed(0)
    d = dev()
    conv = L.Conv2d((H, W, OF.cpad_for(Cin // groups) * groups), K, R,
                    stride=stride, pad=pad, groups=groups, relu=relu,
                    device=str(d), batch=B)
    cphys = conv.Cphys_in
    x = torch.randn(B, H, W, cphys, device=d)
    x[..., :] = q(x)
    x_l = x[..., :Cin] if groups == 1 else x  # grouped: slices are phys
    # logical weight currently in the layer:
    wl = conv.plan.flat_to_w(conv.weight.data).cpu()[:, :Cin // groups]
    bl = conv._get_b().cpu()

    xg = x.to(torch.bfloat16).requires_grad_(False)
    xg.requires_grad_()
    y = conv(xg)
    # reference
    if groups == 1:
        xr = q(x[..., :Cin].cpu()).requires_grad_()
    else:
        cg_p = cphys // groups
        cg_l = Cin // groups
        parts = [x[..., g * cg_p:g * cg_p + cg_l] for g in range(groups)]
        xr = q(torch.cat(parts, dim=-1).cpu()).requires_grad_()
    yr = OF.conv2d_ref(xr, q(wl), bl.float(), (stride, stride), (pad, pad),
                       groups, relu)
    got = y.float().cpu()[..., [g * conv.plan.Kgp + k for g in range(groups)
                                for k in range(conv.plan.Kg)]]
    e_fwd = rel_err(got, yr)
    assert e_fwd < atol, f"fwd rel err {e_fwd}"

    # backward
    dy = torch.randn_like(yr)
    dy = q(dy)
    dyg = torch.zeros(y.shape, device=d)
    for g in range(groups):
        dyg[..., g * conv.plan.Kgp:g * conv.plan.Kgp + conv.plan.Kg] = \
            dy[..., g * conv.plan.Kg:(g + 1) * conv.plan.Kg].to(d)
    y.backward(dyg.to(torch.bfloat16))
    yr.backward(dy)

    # dW: map layer grad (flat) back to logical
    dw_flat = conv.weight.grad
    dw_log = conv.plan.flat_to_w(dw_flat).cpu()[:, :Cin // groups]
    # reference autograd grad wrt logical weight:
    # compute by autograd on wl
    xr2 = xr.detach()
    wl2 = q(wl).requires_grad_()
    yr2 = OF.conv2d_ref(xr2, wl2, bl.float(), (stride, stride), (pad, pad),
                        groups, relu)
    yr2.backward(dy)
    e_wg = rel_err(dw_log, wl2.grad)
    assert e_wg < atol, f"wgrad rel err {e_wg}"
    # bias grad: gather padded layout
    dbg = torch.cat([conv.bias.grad[g * conv.plan.Kgp:
                                    g * conv.plan.Kgp + conv.plan.Kg]
                     for g in range(groups)]).cpu()
    ref_db = dy.sum(dim=(0, 1, 2))
    if relu:
        ref_db = None  # bias grad folded through relu; compare via wl2
    if ref_db is not None:
        assert rel_err(dbg, ref_db) < atol
    # dx
    if groups == 1:
        dx_got = xg.grad.float().cpu()[..., :Cin]
    else:
        cg_p = cphys // groups
        cg_l = Cin // groups
        parts = [xg.grad.float().cpu()[..., g * cg_p:g * cg_p + cg_l]
                 for g in range(groups)]
        dx_got = torch.cat(parts, dim=-1)
    e_dx = rel_err(dx_got, xr.grad)
    assert e_dx < atol, f"dgrad rel err {e_dx}"
    print(f"[margins atol={atol}] fwd {e_fwd:.4f} wgrad {e_wg:.4f} "
          f"dgrad {e_dx:.4f}", flush=True)


def test_conv_3x3_s1():
    conv_case(4, 12, 12, 32, 64, 3, 1, 1)


def test_conv_3x3_s1_relu():
    conv_case(4, 12, 12, 32, 64, 3, 1, 1, relu=True)


def test_conv_1x1():
    conv_case(4, 9, 9, 64, 32, 1, 1, 0)


def test_conv_stem_11x11_s4():
    # stem layers run with C_phys = 4 (RGB + one zero pad channel fed by
    # the loader); the kernel is exercised on all 4 channels here.
    # Wider tier than the 2e-2 default: a 121-tap dot over only 4
    # channels cancels heavily, so max-normalized error lands ~3.5e-2
    # from bf16 input quantization alone (measured stable across
    # rounds; the variants cross-check pins kernel-vs-kernel to 5e-3).
    conv_case(2, 83, 83, 4, 96, 11, 4, 0, atol=0.05)


def test_conv_5x5_grouped():
    conv_case(2, 13, 13, 96, 64, 5, 1, 2, groups=2)


def test_conv_3x3_s2():
    conv_case(2, 14, 14, 32, 48, 3, 2, 1)


def test_conv_7x7_s2_p3():
    conv_case(2, 28, 28, 16, 32, 7, 2, 3)


def test_linear():
    from theanompi_amd.models import layers as L
    torch.manual_seed(1)
    d = dev()
    lin = L.Linear(200, 75, device=str(d), batch=8)
    x = q(torch.randn(8, 200))
    wl = lin.weight.data[0, :75, :200].t().cpu().clone()
    bl = lin.bias.data[:75].cpu().clone()
    xg = x.to(d).to(torch.bfloat16).requires_grad_()
    y = lin(xg)
    xr = x.clone().requires_grad_()
    yr = xr @ q(wl) + bl
    assert rel_err(y.float().cpu()[:, :75], yr) < 0.05
    dy = q(torch.randn_like(yr))
    dyg = torch.zeros(y.shape, device=d)
    dyg[:, :75] = dy.to(d)
    y.backward(dyg.to(torch.bfloat16))
    yr.backward(dy)
    dw_log = lin.weight.grad[0, :75, :200].t().cpu()
    wl2 = q(wl).requires_grad_()
    yr2 = xr.detach() @ wl2 + bl
    yr2.backward(dy)
    assert rel_err(dw_log, wl2.grad) < 0.05
    assert rel_err(xg.grad.float().cpu()[:, :200], xr.grad) < 0.05


# ---------------------------------------------------------------------------
# stateless ops
# ---------------------------------------------------------------------------

def test_maxpool():
    from theanompi_amd.ops import functional as OF
    torch.manual_seed(2)
    x = q(torch.randn(2, 13, 13, 32))
    xg = x.to(dev()).to(torch.bfloat16).requires_grad_()
    y = OF.maxpool2d(xg, 3, 2)
    xr = x.clone().requires_grad_()
    yr = OF.maxpool2d_ref(xr, (3, 3), (2, 2), (0, 0))
    assert rel_err(y, yr) < 1e-2
    dy = q(torch.randn_like(yr))
    y.backward(dy.to(dev()).to(torch.bfloat16))
    yr.backward(dy)
    assert rel_err(xg.grad, xr.grad) < 2e-2


def test_maxpool_padded():
    from theanompi_amd.ops import functional as OF
    x = q(torch.randn(2, 8, 8, 16))
    xg = x.to(dev()).to(torch.bfloat16).requires_grad_()
    y = OF.maxpool2d(xg, 3, 2, 1)
    xr = x.clone().requires_grad_()
    yr = OF.maxpool2d_ref(xr, (3, 3), (2, 2), (1, 1))
    assert rel_err(y, yr) < 1e-2
    dy = q(torch.randn_like(yr))
    y.backward(dy.to(dev()).to(torch.bfloat16))
    yr.backward(dy)
    assert rel_err(xg.grad, xr.grad) < 2e-2


def test_avgpool():
    from theanompi_amd.ops import functional as OF
    x = q(torch.randn(2, 12, 12, 16))
    xg = x.to(dev()).to(torch.bfloat16).requires_grad_()
    y = OF.avgpool2d(xg, 3, 2, 1)
    xr = x.clone().requires_grad_()
    yr = OF.avgpool2d_ref(xr, (3, 3), (2, 2), (1, 1))
    assert rel_err(y, yr) < 2e-2
    dy = q(torch.randn_like(yr))
    y.backward(dy.to(dev()).to(torch.bfloat16))
    yr.backward(dy)
    assert rel_err(xg.grad, xr.grad) < 2e-2


def test_global_avgpool():
    from theanompi_amd.ops import functional as OF
    x = q(torch.randn(2, 7, 7, 64))
    xg = x.to(dev()).to(torch.bfloat16)
    y = OF.avgpool2d(xg, 7, 7, 0)
    ref = x.mean(dim=(1, 2), keepdim=True)
    assert rel_err(y, ref) < 2e-2


def test_lrn():
    from theanompi_amd.ops import functional as OF
    torch.manual_seed(3)
    x = q(torch.randn(2, 6, 6, 32))
    xg = x.to(dev()).to(torch.bfloat16).requires_grad_()
    y = OF.lrn(xg, 5, 1e-4, 0.75, 1.0)
    xr = x.clone().requires_grad_()
    yr = OF.lrn_ref(xr, 5, 1e-4, 0.75, 1.0)
    assert rel_err(y, yr) < 2e-2
    dy = q(torch.randn_like(yr))
    y.backward(dy.to(dev()).to(torch.bfloat16))
    yr.backward(dy)
    assert rel_err(xg.grad, xr.grad) < 3e-2
    # C % 16 != 0 exercises the 4-wide fallback kernels
    x8 = q(torch.randn(2, 6, 6, 8))
    xg8 = x8.to(dev()).to(torch.bfloat16).requires_grad_()
    y8 = OF.lrn(xg8, 5, 1e-4, 0.75, 1.0)
    xr8 = x8.clone().requires_grad_()
    yr8 = OF.lrn_ref(xr8, 5, 1e-4, 0.75, 1.0)
    assert rel_err(y8, yr8) < 2e-2
    dy8 = q(torch.randn_like(yr8))
    y8.backward(dy8.to(dev()).to(torch.bfloat16))
    yr8.backward(dy8)
    assert rel_err(xg8.grad, xr8.grad) < 3e-2


def test_batchnorm_train_and_eval():
    from theanompi_amd.ops import functional as OF
    torch.manual_seed(4)
    x = q(torch.randn(8, 6, 6, 32) * 2 + 0.5)
    gamma = torch.randn(32).abs() + 0.5
    beta = torch.randn(32) * 0.1
    d = dev()
    rm_g = torch.zeros(32, device=d)
    rv_g = torch.ones(32, device=d)
    rm_c = torch.zeros(32)
    rv_c = torch.ones(32)
    gg = gamma.to(d).requires_grad_()
    bg = beta.to(d).requires_grad_()
    xg = x.to(d).to(torch.bfloat16).requires_grad_()
    y = OF.batchnorm(xg, gg, bg, rm_g, rv_g, True, 0.1, 1e-5)
    gr = gamma.clone().requires_grad_()
    br = beta.clone().requires_grad_()
    xr = x.clone().requires_grad_()
    yr = OF.batchnorm_ref(xr, gr, br, rm_c, rv_c, True, 0.1, 1e-5)
    assert rel_err(y, yr) < 3e-2
    assert rel_err(rm_g.cpu(), rm_c) < 3e-2
    assert rel_err(rv_g.cpu(), rv_c) < 3e-2
    dy = q(torch.randn_like(yr))
    y.backward(dy.to(d).to(torch.bfloat16))
    yr.backward(dy)
    assert rel_err(xg.grad, xr.grad) < 5e-2
    assert rel_err(gg.grad.cpu(), gr.grad) < 5e-2
    assert rel_err(bg.grad.cpu(), br.grad) < 5e-2
    # eval mode
    with torch.no_grad():
        ye = OF.batchnorm(xg.detach(), gg.detach(), bg.detach(), rm_g,
                          rv_g, False, 0.1, 1e-5)
        yre = OF.batchnorm_ref(x, gamma, beta, rm_c, rv_c, False, 0.1,
                               1e-5)
    assert rel_err(ye, yre) < 3e-2


def test_dropout():
    from theanompi_amd.ops import functional as OF
    x = torch.ones(100000, 16, device=dev(), dtype=torch.bfloat16)
    xg = x.requires_grad_()
    y = OF.dropout(xg, 0.4, True)
    keep = (y.float() != 0).float().mean().item()
    assert abs(keep - 0.6) < 0.01
    kept_vals = y.float()[y.float() != 0]
    assert torch.allclose(kept_vals,
                          torch.full_like(kept_vals, 1 / 0.6), atol=5e-2)
    y.backward(torch.ones_like(y))
    g = xg.grad.float()
    assert torch.equal((g != 0), (y.float() != 0))


def test_softmax_xent():
    from theanompi_amd.ops import functional as OF
    torch.manual_seed(5)
    B, CLS, CP = 32, 100, 112
    logits = q(torch.randn(B, CP) * 3)
    labels = torch.randint(0, CLS, (B,))
    lg = logits.to(dev()).to(torch.bfloat16).requires_grad_()
    loss, probs = OF.softmax_xent(lg, labels.to(dev()), CLS)
    lr_ = logits.clone().requires_grad_()
    loss_r, probs_r = OF.softmax_xent_ref(lr_, labels, CLS)
    assert abs(float(loss) - float(loss_r)) < 2e-2 * max(1, float(loss_r))
    assert rel_err(probs[:, :CLS], probs_r[:, :CLS]) < 3e-2
    loss.backward()
    loss_r.backward()
    assert rel_err(lg.grad.float().cpu()[:, :CLS], lr_.grad[:, :CLS]) < 3e-2
    assert lg.grad.float()[:, CLS:].abs().sum() == 0
    # fused loss-mean + top-1 err (kernel atomics) vs torch reductions
    lg2 = logits.to(dev()).to(torch.bfloat16)
    loss_f, _p, err_f = OF.softmax_xent_err(lg2, labels.to(dev()), CLS)
    pred = lg2.float()[:, :CLS].argmax(dim=1)
    err_ref = (pred != labels.to(dev())).float().mean()
    assert abs(float(err_f) - float(err_ref)) < 1e-6
    assert abs(float(loss_f) - float(loss_r)) < 2e-2 * max(1, float(loss_r))


def test_sgd_step_kernel():
    from theanompi_amd import ops as O
    C = O.ext()
    torch.manual_seed(6)
    n = 1003
    d = dev()
    w = torch.randn(n, device=d)
    g = torch.randn(n, device=d)
    v = torch.randn(n, device=d)
    out = torch.empty(n, dtype=torch.bfloat16, device=d)
    w0, g0, v0 = w.clone(), g.clone(), v.clone()
    C.sgd_step(w, g, v, out, 0.1, 0.9, 1e-4, 0)
    vr = 0.9 * v0 - 0.1 * (g0 + 1e-4 * w0)
    wr = w0 + vr
    assert rel_err(v, vr) < 1e-5
    assert rel_err(w, wr) < 1e-5
    assert torch.equal(out, wr.to(torch.bfloat16))
    # phase1/phase2 compose to the same result
    w2, g2, v2 = w0.clone(), g0.clone(), v0.clone()
    empty = torch.empty(0, dtype=torch.bfloat16, device=d)
    C.sgd_step(w2, g2, v2, empty, 0.1, 0.9, 1e-4, 1)
    C.sgd_step(w2, v2, torch.zeros_like(v2), empty, 0.1, 0.9, 1e-4, 2)
    assert rel_err(w2, wr) < 1e-5


def test_conv_variants_numerics():
    """Every tuned kernel variant computes the same conv as variant 0
    (fwd within bf16 rounding of each other; wgrad fp32 within
    accumulation-order noise)."""
    from theanompi_amd.models import layers as L
    torch.manual_seed(11)
    d = dev()
    # M=8192 -> 64 m-tiles (%8==0) and 2 n-tiles at BN=128 so the
    # NFAST variants (11-13) take their real path instead of falling
    # back to the m-fast twins
    B, H, W, Cin, K = 8, 32, 32, 128, 256
    conv = L.Conv2d((H, W, Cin), K, 3, stride=1, pad=1, relu=False,
                    device=str(d), batch=B)
    x = torch.randn(B, H, W, Cin, device=d).to(torch.bfloat16)
    x.requires_grad_()
    dy = torch.randn(B, H, W, conv.plan.Kphys, device=d) \
        .to(torch.bfloat16)
    M = B * conv.plan.P * conv.plan.Q

    def run(vf, vw):
        conv._v_fwd, conv._v_M, conv._v_wgrad = vf, M, (vw, 0)
        for p2 in conv.plan.dgrad_plans:
            if p2 is not None:
                p2["variant"], p2["vN"] = vf, B
        x.grad = None
        conv.weight.grad = None
        conv.bias.grad = None
        y = conv(x)
        y.backward(dy)
        return (y.detach().float(), conv.weight.grad.clone(),
                x.grad.float().clone())

    y0, dw0, dx0 = run(0, 0)
    for vf in (1, 2, 3, 5, 6, 7, 8, 9, 10, 11, 12, 13):
        y, _dw, dx = run(vf, 0)
        assert rel_err(y, y0) < 5e-3, (vf, rel_err(y, y0))
        assert rel_err(dx, dx0) < 5e-3, (vf, rel_err(dx, dx0))
    for vw in (1, 2, 3, 4, 5, 6, 7, 8):
        _y, dw, _dx = run(0, vw)
        assert rel_err(dw, dw0) < 5e-3, (vw, rel_err(dw, dw0))


def test_googlenet_checkpoint_roundtrip(tmp_path):
    """GPU GoogLeNet (merged inception) save -> clobber -> load
    restores every logical weight AND velocity bit-exactly — the
    reference checkpoint layout survives the merged layers."""
    from theanompi_amd.models.googlenet import GoogLeNet
    torch.manual_seed(2)
    m = GoogLeNet({"device": "cuda:0", "batch_size": 4, "n_iters": 1,
                   "val_iters": 1})
    with torch.no_grad():
        for v in m.vels:
            v.normal_()                       # nonzero SGD state
    plist = m._logical_param_list()
    before = [g().detach().float().cpu().clone() for (_n, g, _s) in
              plist]
    with m._vels_as_params():
        vbefore = [g().detach().float().cpu().clone()
                   for (_n, g, _s) in plist]
    m.save(str(tmp_path), 1)
    with torch.no_grad():
        for p in m.params:
            p.normal_()
        for v in m.vels:
            v.zero_()
    m.load(str(tmp_path), 1)
    after = [g().detach().float().cpu() for (_n, g, _s) in plist]
    with m._vels_as_params():
        vafter = [g().detach().float().cpu() for (_n, g, _s) in plist]
    for i, (a, b) in enumerate(zip(before, after)):
        assert torch.equal(a, b), f"weight {i}"
    for i, (a, b) in enumerate(zip(vbefore, vafter)):
        assert torch.equal(a, b), f"vel {i}"


def test_batchnorm_fused_residual():
    """y = relu(bn(x) + res) in one pass vs a manual fp32 reference
    that shares the fused path's own relu mask — forward and ALL grads
    (dx, dgamma, dbeta, dres).  (Comparing against the separate
    bn+add_relu composition is ill-posed at relu boundaries: its extra
    bf16 rounding flips borderline masks, and one flipped position
    shifts max-normalized dx by ~0.7 — measured, both paths match the
    fp32 reference on non-borderline elements to <0.3%.)"""
    from theanompi_amd.models import layers as L
    torch.manual_seed(9)
    d = dev()
    N, H, W, C = 8, 14, 14, 64
    M = N * H * W
    bn1 = L.BatchNorm2d(C, relu=False, device=str(d))
    x0 = torch.randn(N, H, W, C, device=d).to(torch.bfloat16)
    r0 = torch.randn(N, H, W, C, device=d).to(torch.bfloat16)
    dy = torch.randn(N, H, W, C, device=d).to(torch.bfloat16)

    x1 = x0.clone().requires_grad_()
    r1 = r0.clone().requires_grad_()
    y1 = bn1(x1, residual=r1, relu=True)          # fused
    y1.backward(dy)

    with torch.no_grad():
        xf = x0.float()
        mean = xf.mean((0, 1, 2))
        var = xf.var((0, 1, 2), unbiased=False)
        invstd = 1.0 / torch.sqrt(var + bn1.eps)
        xhat = (xf - mean) * invstd
        yref = torch.relu(xhat + r0.float())
        g = dy.float() * (y1.float() > 0)         # the FUSED mask
        s1 = g.sum((0, 1, 2))
        s2 = (g * xhat).sum((0, 1, 2))
        dxref = invstd * (g - s1 / M - xhat * s2 / M)
    assert rel_err(y1, yref) < 1.5e-2
    assert rel_err(x1.grad, dxref) < 5e-3
    assert rel_err(r1.grad, g) < 1e-6              # exact masked dy
    assert rel_err(bn1.gamma.grad, s2) < 1e-5
    assert rel_err(bn1.beta.grad, s1) < 1e-5


def test_inception_merged_equivalence():
    """GPU inception (fused 3-way 1x1 GEMM + concat-writing branches)
    matches the unmerged CPU reference given identical weights loaded
    through the logical checkpoint mapping; branch weight grads
    match too."""
    from theanompi_amd.models.googlenet import _Inception
    torch.manual_seed(3)
    B, hw, cin = 4, 14, 64          # widths all multiples of 16: no pad
    args = (hw, cin, 32, 48, 64, 16, 32, 32)
    gi = _Inception(*args, dev(), B)
    ci = _Inception(*args, "cpu", B)
    assert gi.merged and not ci.merged
    gp, cp = gi.logical_params(), ci.logical_params()
    assert len(gp) == len(cp)
    for (_n1, g1, _s1), (_n2, _g2, s2) in zip(gp, cp):
        s2(g1().float().cpu())       # GPU weights -> CPU model
    x = q(torch.randn(B, hw, hw, cin))
    yg = gi(x.to(dev()).to(torch.bfloat16).requires_grad_())
    yc = ci(x.clone().requires_grad_())
    assert yg.shape == yc.shape
    assert rel_err(yg, yc) < 2e-2

    # gradient check vs STANDALONE GPU 1x1 convs (identical kernels and
    # bf16 relu decisions — the CPU fp32 baseline flips borderline relu
    # masks and diverges in wgrad; forward above pins CPU parity)
    from theanompi_amd.models import layers as L
    ks = [32, 48, 16]
    singles = []
    for bi, k in enumerate(ks):
        c = L.Conv2d((hw, hw, cin), k, 1, relu=True, device=str(dev()),
                     batch=B)
        off = gi.b135.branch_off[bi]
        with torch.no_grad():
            c.weight.data.copy_(
                gi.b135.weight.data[:, off:off + k, :])
            c.bias.data.copy_(gi.b135.bias.data[off:off + k])
        c.post_step()
        singles.append(c)
    xg2 = x.to(dev()).to(torch.bfloat16).requires_grad_()
    m = gi.b135(xg2)
    dy = torch.randn_like(m)
    m.backward(dy)
    xg3 = x.to(dev()).to(torch.bfloat16).requires_grad_()
    for bi, c in enumerate(singles):
        off = gi.b135.branch_off[bi]
        kp = gi.b135.branch_kp[bi]
        ys = c(xg3)
        ys.backward(dy[..., off:off + kp])
        gm = gi.b135.weight.grad[0, off:off + kp, :]
        assert rel_err(gm, c.weight.grad[0]) < 1e-3, bi
        bm = gi.b135.bias.grad[off:off + kp]
        assert rel_err(bm, c.bias.grad) < 1e-3, bi
    # fused dx sum: the STANDALONE baseline rounds each branch's dx to
    # bf16 and accumulates in bf16 (3 roundings + bf16 adds); the
    # merged GEMM accumulates everything in fp32 with one rounding, so
    # they differ by baseline rounding noise (~5e-3 measured)
    assert rel_err(xg2.grad, xg3.grad) < 2e-2


def test_sgd_cached_descriptor_lr_invalidate():
    """adjust_hyperp mid-run: the persistent-grads cached descriptor
    must be rebuilt after opt.invalidate() so the new LR takes effect
    (VERDICT r1 weak #8)."""
    from theanompi_amd.lib.opt import ParamRec, SGDMomentum
    torch.manual_seed(7)
    d = dev()
    masters = [torch.randn(257, device=d, requires_grad=True),
               torch.randn(64, 3, device=d, requires_grad=True)]
    recs = [ParamRec(p, f"p{i}") for i, p in enumerate(masters)]
    opt = SGDMomentum(recs, lr=0.1, momentum=0.9, weight_decay=0.0)
    opt.make_persistent()
    ref_w = [p.detach().clone() for p in masters]
    ref_v = [torch.zeros_like(p) for p in masters]

    def ref_step(lr):
        for i in range(len(ref_w)):
            ref_v[i] = 0.9 * ref_v[i] - lr * grads[i]
            ref_w[i] += ref_v[i]

    grads = [torch.randn_like(p) for p in masters]
    for r, g in zip(recs, grads):
        r.master.grad.copy_(g)
    opt.step()
    ref_step(0.1)
    opt.lr = 0.01            # adjust_hyperp path
    opt.invalidate()
    opt.step()
    ref_step(0.01)
    for r, w in zip(recs, ref_w):
        assert rel_err(r.master.data, w) < 1e-5


def test_gather_bf16():
    from theanompi_amd import ops as O
    C = O.ext()
    src = torch.randn(100, device=dev()).to(torch.bfloat16)
    idx = torch.tensor([0, 5, 99, -1, 7], dtype=torch.int32, device=dev())
    dst = torch.empty(5, dtype=torch.bfloat16, device=dev())
    C.gather_bf16(src, idx, dst)
    assert torch.equal(dst[:3].float(),
                       src[[0, 5, 99]].float())
    assert float(dst[3]) == 0.0
    assert float(dst[4]) == float(src[7])


# ---------------------------------------------------------------------------
# end-to-end: tiny model trains on GPU
# ---------------------------------------------------------------------------

def test_testmodel_gpu_trains():
    from theanompi_amd.models.test_model import TestModel
    torch.manual_seed(0)
    m = TestModel({"rank": 0, "size": 1, "device": "cuda:0",
                   "verbose": False})
    m.compile_iter_fns("grad")
    losses = []
    for i in range(30):
        loss, err = m.train_iter(i % m.data.n_batch_train)
        m.apply_update()
        losses.append(float(loss))
    assert losses[-1] < losses[0], losses[:3] + losses[-3:]


@pytest.mark.gpu
@pytest.mark.parametrize("relu,tiny", [(True, False), (False, False),
                                       (False, True), (True, True)])
def test_batchnorm_bwd_xhat_from_y(relu, tiny):
    """BN backward reconstructs xhat from the saved bf16 output when all
    |gamma| are safe (skipping the x read); a tiny gamma must fall back
    to the exact x path.  Both paths vs the fp32 reference."""
    from theanompi_amd.ops import functional as OF
    torch.manual_seed(9)
    x = q(torch.randn(16, 7, 7, 64) * 1.5 - 0.2)
    gamma = torch.randn(64).abs() + 0.5
    if tiny:
        gamma[3] = 1e-4                 # trips the allok=0 fallback
    beta = torch.randn(64) * 0.3
    d = dev()
    gg = gamma.to(d).requires_grad_()
    bg = beta.to(d).requires_grad_()
    xg = x.to(d).to(torch.bfloat16).requires_grad_()
    rm_g, rv_g = torch.zeros(64, device=d), torch.ones(64, device=d)
    y = OF.batchnorm(xg, gg, bg, rm_g, rv_g, True, 0.1, 1e-5, relu=relu)
    gr = gamma.clone().requires_grad_()
    br = beta.clone().requires_grad_()
    xr = x.clone().requires_grad_()
    yr = OF.batchnorm_ref(xr, gr, br, torch.zeros(64), torch.ones(64),
                          True, 0.1, 1e-5, relu=relu)
    dy = q(torch.randn_like(yr))
    y.backward(dy.to(d).to(torch.bfloat16))
    yr.backward(dy)
    assert rel_err(xg.grad, xr.grad) < 5e-2
    assert rel_err(gg.grad.cpu(), gr.grad) < 5e-2
    assert rel_err(bg.grad.cpu(), br.grad) < 5e-2


@pytest.mark.gpu
def test_relu_bwd_sliced_dy():
    """relu_bwd consumes a channel slice of a concat buffer directly
    (the inception backward split) — no .contiguous() copy."""
    from theanompi_amd import ops as _ops
    C = _ops.ext()
    d = dev()
    torch.manual_seed(11)
    full = q(torch.randn(4, 6, 6, 96)).to(d).to(torch.bfloat16)
    y = q(torch.randn(4, 6, 6, 32)).to(d).to(torch.bfloat16)
    sl = full[..., 32:64]                   # non-contiguous channel slice
    assert not sl.is_contiguous()
    out = torch.empty(sl.shape, dtype=torch.bfloat16, device=d)
    C.relu_bwd(sl, y, out)
    ref = torch.where(y.float() > 0, sl.float(), torch.zeros(()).to(d))
    assert rel_err(out, ref) == 0.0 or rel_err(out, ref) < 1e-6


@pytest.mark.gpu
def test_inception_concat_write_equivalence():
    """Concat-writing inception (branches write channel slices of the
    concat buffer in place, zero-copy split in backward) must match the
    torch.cat composition exactly — forward, input grad, param grads."""
    import torch as T
    from theanompi_amd.models.googlenet import _Inception
    T.manual_seed(5)
    d = dev()
    m = _Inception(8, 32, 16, 16, 32, 16, 16, 16, str(d), 4)
    x = q(T.randn(4, 8, 8, 32)).to(d).to(T.bfloat16)

    def run(fn):
        xi = x.clone().requires_grad_()
        y = fn(xi)
        loss = (y.float() ** 2).mean()
        loss.backward()
        grads = [p.grad.detach().clone() for p in m.parameters()]
        for p in m.parameters():
            p.grad = None
        return y.detach().float(), xi.grad.detach().float(), grads

    def old_path(xi):
        mm = m.b135(xi)
        s1, s3, s5 = m.b135.slices(mm)
        return T.cat([s1, m.b3b(s3), m.b5b(s5),
                      m.bp(m.pool(xi))], dim=-1)

    y_new, dx_new, g_new = run(m)               # concat-write path
    y_old, dx_old, g_old = run(old_path)
    assert T.equal(y_new, y_old)
    assert rel_err(dx_new, dx_old.cpu()) < 1e-6
    for a, b in zip(g_new, g_old):
        assert rel_err(a, b.cpu()) < 2e-2       # wgrad atomics reorder


@pytest.mark.gpu
def test_conv_fused_bn_stats():
    """The conv epilogue's bucketed sum/sum-sq must equal torch sums
    over the produced bf16 output (fused BN statistics)."""
    import torch as T
    from theanompi_amd.models import layers as L
    T.manual_seed(12)
    d = dev()
    # shape must stay off the split-K path (which correctly declines
    # to collect): M = 8*56*56 = 25088 -> ~196 base blocks
    conv = L.Conv2d((56, 56, 64), 96, 3, pad=1, relu=False,
                    device=str(d), batch=8, collect_stats=True,
                    has_bias=False)
    conv.train()
    x = q(T.randn(8, 56, 56, 64)).to(d).to(T.bfloat16)
    y = conv(x)
    st = conv.pop_stats()
    assert st is not None
    C = y.shape[-1]
    got = st.view(32, 2, C).sum(dim=0)
    yf = y.detach().float().reshape(-1, C)
    ref1 = yf.sum(dim=0)
    ref2 = (yf * yf).sum(dim=0)
    assert rel_err(got[0], ref1.cpu()) < 1e-4
    assert rel_err(got[1], ref2.cpu()) < 1e-4
    # buckets are consumed by the BN side; a second forward refills
    y2 = conv(x)
    st2 = conv.pop_stats()
    assert st2 is not None


@pytest.mark.gpu
def test_convbn_fused_stats_equivalence():
    """_ConvBN with fused conv-epilogue statistics matches the separate
    bn_sums path (forward + grads)."""
    import torch as T
    from theanompi_amd.models import resnet50 as R
    T.manual_seed(13)
    d = dev()

    def build(fuse):
        old = R._BN_FUSE
        R._BN_FUSE = fuse
        try:
            T.manual_seed(13)
            m = R._ConvBN(56, 32, 48, 3, pad=1, relu=True, dev=str(d),
                          B=8)      # off the split-K path
        finally:
            R._BN_FUSE = old
        m.train()
        return m

    ma, mb = build(True), build(False)
    mb.load_state_dict(ma.state_dict())
    # the fused path must actually engage
    assert ma.conv._collect_stats
    x = q(T.randn(8, 56, 56, 32)).to(d).to(T.bfloat16)

    def run(m):
        xi = x.clone().requires_grad_()
        y = m(xi)
        (y.float() ** 2).mean().backward()
        gs = [p.grad.detach().clone() for p in m.parameters()]
        for p in m.parameters():
            p.grad = None
        return y.detach().float(), xi.grad.detach().float(), gs

    ya, dxa, ga = run(ma)
    yb, dxb, gb = run(mb)
    # the two paths accumulate the channel statistics through DIFFERENT
    # atomic orderings (conv-epilogue buckets vs bn_sums blocks), so
    # mean/invstd differ by fp32 rounding and y by a few bf16 ulp —
    # measured up to ~3.5e-3 depending on scheduling (box/run dependent)
    assert rel_err(ya, yb.cpu()) < 8e-3
    assert rel_err(dxa, dxb.cpu()) < 2e-2
    for a, b in zip(ga, gb):
        assert rel_err(a, b.cpu()) < 2e-2
