"""Isolate the fused-residual BN backward dx discrepancy."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch  # noqa: E402

from theanompi_amd.models import layers as L  # noqa: E402
from theanompi_amd.ops import functional as OF  # noqa: E402

torch.manual_seed(9)
d = torch.device("cuda", 0)
N, H, W, C = 8, 14, 14, 64
bn1 = L.BatchNorm2d(C, relu=False, device=str(d))
x0 = torch.randn(N, H, W, C, device=d).to(torch.bfloat16)
r0 = torch.randn(N, H, W, C, device=d).to(torch.bfloat16)
dy = torch.randn(N, H, W, C, device=d).to(torch.bfloat16)

x1 = x0.clone().requires_grad_()
r1 = r0.clone().requires_grad_()
y1 = bn1(x1, residual=r1, relu=True)
y1.backward(dy)

# manual x-path reference in fp32
with torch.no_grad():
    xf = x0.float()
    M = N * H * W
    mean = xf.mean((0, 1, 2))
    var = xf.var((0, 1, 2), unbiased=False)
    invstd = 1.0 / torch.sqrt(var + bn1.eps)
    xhat = (xf - mean) * invstd
    yref = torch.relu(xhat + r0.float())
    print("fwd err:", (y1.float() - yref).abs().max().item())
    g = dy.float() * (yref > 0)
    s1 = g.sum((0, 1, 2))
    s2 = (g * xhat).sum((0, 1, 2))
    dxref = invstd * (g - s1 / M - xhat * s2 / M)
    e = (x1.grad.float() - dxref).abs().max() / dxref.abs().max()
    print("dx err:", e.item())
    print("dres err:",
          ((r1.grad.float() - g).abs().max()
           / g.abs().max().clamp_min(1e-6)).item())
    print("dgamma err:",
          ((bn1.gamma.grad - s2).abs().max()
           / s2.abs().max()).item())
    print("dbeta err:",
          ((bn1.beta.grad - s1).abs().max() / s1.abs().max()).item())
    # per-position diagnosis of dx
    dxg = x1.grad.float()
    bad = (dxg - dxref).abs() > 0.05 * dxref.abs().max()
    print("bad frac:", bad.float().mean().item())
    # is dx missing the mean-sub terms?
    alt = invstd * g
    print("err if dx==invstd*g:",
          ((dxg - alt).abs().max() / dxref.abs().max()).item())
    # or unmasked g?
    gu = dy.float()
    s1u = gu.sum((0, 1, 2))
    s2u = (gu * xhat).sum((0, 1, 2))
    altu = invstd * (gu - s1u / M - xhat * s2u / M)
    print("err if unmasked:",
          ((dxg - altu).abs().max() / dxref.abs().max()).item())
    # or y-path xhat reconstruction from fused y?
    xh_y = yref            # gamma=1, beta=0: (y-beta)/gamma = y
    s2y = (g * xh_y).sum((0, 1, 2))
    alty = invstd * (g - s1 / M - xh_y * s2y / M)
    print("err if xhat-from-fused-y:",
          ((dxg - alty).abs().max() / dxref.abs().max()).item())

# --- separate composition vs the same manual reference --------------------
bn2 = L.BatchNorm2d(C, relu=False, device=str(d))
x2 = x0.clone().requires_grad_()
r2 = r0.clone().requires_grad_()
y2 = OF.add(bn2(x2), r2, relu=True)
y2.backward(dy)
with torch.no_grad():
    print("sep fwd err:", (y2.float() - yref).abs().max().item())
    e2 = (x2.grad.float() - dxref).abs().max() / dxref.abs().max()
    print("sep dx err:", e2.item())
    print("sep dbeta err:",
          ((bn2.beta.grad - s1).abs().max() / s1.abs().max()).item())
    print("sep dgamma err:",
          ((bn2.gamma.grad - s2).abs().max() / s2.abs().max()).item())
    # y-path reconstruction uses bn2's own bf16 output:
    ybn = (xhat).to(torch.bfloat16).float()
    s2yy = (g * ybn).sum((0, 1, 2))
    altyy = invstd * (g - s1 / M - ybn * s2yy / M)
    print("sep err if xhat-from-own-y:",
          ((x2.grad.float() - altyy).abs().max()
           / dxref.abs().max()).item())

# --- state dependence probes ----------------------------------------------
import subprocess
import json

with torch.no_grad():
    big = dxref.abs() > 0.1 * dxref.abs().max()
    ratio = (x2.grad.float()[big] / dxref[big])
    print("sep dx/ref ratio: med", ratio.median().item(),
          "q10", ratio.quantile(0.1).item(),
          "q90", ratio.quantile(0.9).item())
    # y-path model with UNMASKED g (in case the add backward is wrong)
    gu2 = dy.float()
    s1u2 = gu2.sum((0, 1, 2))
    s2u2 = (gu2 * ybn).sum((0, 1, 2))
    altyu = invstd * (gu2 - s1u2 / M - ybn * s2u2 / M)
    print("sep err if ypath-unmasked:",
          ((x2.grad.float() - altyu).abs().max()
           / dxref.abs().max()).item())

# fresh-process separate-only run (no fused call first)
code = """
import torch, sys
sys.path.insert(0, '.')
from theanompi_amd.models import layers as L
from theanompi_amd.ops import functional as OF
torch.manual_seed(9)
d = torch.device('cuda', 0)
N, H, W, C = 8, 14, 14, 64
bn = L.BatchNorm2d(C, relu=False, device=str(d))
x0 = torch.randn(N, H, W, C, device=d).to(torch.bfloat16)
r0 = torch.randn(N, H, W, C, device=d).to(torch.bfloat16)
dy = torch.randn(N, H, W, C, device=d).to(torch.bfloat16)
x = x0.clone().requires_grad_()
r = r0.clone().requires_grad_()
y = OF.add(bn(x), r, relu=True)
y.backward(dy)
xf = x0.float(); M = N*H*W
mean = xf.mean((0,1,2)); var = xf.var((0,1,2), unbiased=False)
invstd = 1.0/torch.sqrt(var + bn.eps)
xhat = (xf - mean) * invstd
yref = torch.relu(xhat + r0.float())
g = dy.float() * (yref > 0)
s1 = g.sum((0,1,2)); s2 = (g*xhat).sum((0,1,2))
dxref = invstd * (g - s1/M - xhat*s2/M)
print('FRESH sep dx err:',
      ((x.grad.float()-dxref).abs().max()/dxref.abs().max()).item())
"""
r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                   text=True)
print(r.stdout.strip(), r.stderr[-300:] if r.returncode else "")
