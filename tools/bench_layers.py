#!/usr/bin/env python3
"""Per-layer conv kernel microbenchmark (fwd / dgrad / wgrad) over the
zoo's real shapes — the measurement harness behind kernel tuning
("measure, don't guess").  Run on an MI355X:

    python tools/bench_layers.py [--model alexnet] [--iters 20]

Prints per-layer ms and effective TFLOP/s (dense conv FLOPs, excluding
channel padding waste, so the number reflects end-to-end usefulness).

Caveat: the bwd column includes host-side autograd/allocation overhead
(kernel-level bwd time is ~5x lower for small layers — check rocprofv3
when tuning); the fwd column is kernel-dominated and reliable.
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

SHAPES = {
    # name: (B, H, W, Cin_logical, K, R, stride, pad, groups)
    "alexnet": [
        ("conv1", 128, 227, 227, 3, 96, 11, 4, 0, 1),
        ("conv2", 128, 27, 27, 96, 256, 5, 1, 2, 2),
        ("conv3", 128, 13, 13, 256, 384, 3, 1, 1, 1),
        ("conv4", 128, 13, 13, 384, 384, 3, 1, 1, 2),
        ("conv5", 128, 13, 13, 384, 256, 3, 1, 1, 2),
    ],
    "resnet50": [
        ("stem7x7", 256, 224, 224, 3, 64, 7, 2, 3, 1),
        ("l1_1x1a", 256, 56, 56, 256, 64, 1, 1, 0, 1),
        ("l1_3x3", 256, 56, 56, 64, 64, 3, 1, 1, 1),
        ("l1_1x1b", 256, 56, 56, 64, 256, 1, 1, 0, 1),
        ("l3_3x3", 256, 14, 14, 256, 256, 3, 1, 1, 1),
        ("l4_3x3", 256, 7, 7, 512, 512, 3, 1, 1, 1),
    ],
    "vgg16": [
        ("c2_3x3", 64, 224, 224, 64, 64, 3, 1, 1, 1),
        ("c4_3x3", 64, 112, 112, 128, 128, 3, 1, 1, 1),
        ("c7_3x3", 64, 56, 56, 256, 256, 3, 1, 1, 1),
        ("c10_3x3", 64, 28, 28, 512, 512, 3, 1, 1, 1),
    ],
}


def bench_one(name, B, H, W, Cin, K, R, stride, pad, groups, iters):
    from theanompi_amd.models import layers as L
    from theanompi_amd.ops.functional import cpad_for
    dev = "cuda:0"
    cphys = cpad_for(Cin // groups) * groups
    conv = L.Conv2d((H, W, cphys), K, R, stride=stride, pad=pad,
                    groups=groups, relu=True, device=dev, batch=B,
                    needs_dgrad=True)
    x = torch.randn(B, H, W, cphys, device=dev).to(torch.bfloat16)
    x.requires_grad_()
    P, Q = conv.plan.P, conv.plan.Q
    flops = 2.0 * B * P * Q * K * (Cin // groups) * R * R
    # warmup
    for _ in range(3):
        y = conv(x)
        y.backward(torch.ones_like(y))
        x.grad = None
        conv.weight.grad = None
    torch.cuda.synchronize()

    t0 = time.perf_counter()
    for _ in range(iters):
        y = conv(x)
    torch.cuda.synchronize()
    t_fwd = (time.perf_counter() - t0) / iters

    dy = torch.randn_like(y)
    t0 = time.perf_counter()
    for _ in range(iters):
        y = conv(x)
        y.backward(dy)
        x.grad = None
        conv.weight.grad = None
    torch.cuda.synchronize()
    t_all = (time.perf_counter() - t0) / iters
    t_bwd = t_all - t_fwd

    print(f"| {name:9s} | {B:4d}x{H}x{W}x{Cin:4d} -> {K:4d} r{R} s{stride} "
          f"| {t_fwd*1e3:7.3f} | {flops/t_fwd/1e12:6.1f} "
          f"| {t_bwd*1e3:7.3f} | {2*flops/t_bwd/1e12:6.1f} |")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="all",
                    choices=["all"] + sorted(SHAPES))
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    keys = sorted(SHAPES) if args.model == "all" else [args.model]
    print("| layer | shape | fwd ms | fwd TF | bwd ms | bwd TF |")
    print("|---|---|---|---|---|---|")
    for k in keys:
        print(f"| **{k}** | | | | | |")
        for row in SHAPES[k]:
            bench_one(*row, args.iters)


if __name__ == "__main__":
    main()
