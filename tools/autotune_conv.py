#!/usr/bin/env python3
"""Per-shape conv kernel autotuner.

Sweeps the gather-GEMM variants (ops/hip/conv.hip table: tile shape,
BK, pipeline depth) and the wgrad (BNF, split-K target) grid over the
zoo's real layer shapes on an MI355X, and writes the winners to a
tuned table consumed by ``theanompi_amd.ops.tuning``:

    python tools/autotune_conv.py --out gpurun_out/conv_tuned.json \
        --report gpurun_out/autotune_report.md [--model all] [--iters 30]

Method: wall-clock medians over --iters timed calls (3 reps).  The fwd
sweep times forward only; dgrad/wgrad sweeps time fwd+bwd with the
other stages pinned at their current winners, so differences isolate
the swept stage.  Winners are stored as explicit variant ids keyed by
exact GEMM shape; unlisted shapes keep the launcher heuristic.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from bench_layers import SHAPES  # noqa: E402  (tools/ sibling)

# GoogLeNet-32b inception mix: representative shapes per
# stage — the small-kernel regime the 1-GPU profile says is
# latency-bound (profiles/googlenet_r1_*.md).
SHAPES["googlenet"] = [
    ("stem7x7", 32, 224, 224, 3, 64, 7, 2, 3, 1),
    ("stem3x3", 32, 56, 56, 64, 192, 3, 1, 1, 1),
    ("i3a_1x1", 32, 28, 28, 192, 64, 1, 1, 0, 1),
    ("i3a_3x3", 32, 28, 28, 96, 128, 3, 1, 1, 1),
    ("i3b_5x5", 32, 28, 28, 32, 96, 5, 1, 2, 1),
    ("i4_1x1", 32, 14, 14, 480, 192, 1, 1, 0, 1),
    ("i4_3x3", 32, 14, 14, 112, 224, 3, 1, 1, 1),
    ("i5_1x1", 32, 7, 7, 832, 256, 1, 1, 0, 1),
    ("i5_3x3", 32, 7, 7, 160, 320, 3, 1, 1, 1),
]
SHAPES["fc"] = [
    ("fc6", 128, 1, 1, 9216, 4096, 1, 1, 0, 1),
    ("fc7", 128, 1, 1, 4096, 4096, 1, 1, 0, 1),
]

FWD_VARIANTS = [1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13]
DGRAD_VARIANTS = [1, 2, 5, 6, 7, 8, 9, 10, 11, 12, 13]
WGRAD_GRID = [(v, t) for v in (1, 2, 3, 4, 5, 6, 7, 8)
              for t in (512, 1024, 2048, 4096)]


def build(name, B, H, W, Cin, K, R, stride, pad, groups):
    from theanompi_amd.models import layers as L
    from theanompi_amd.ops.functional import cpad_for
    dev = "cuda:0"
    cphys = cpad_for(Cin // groups) * groups
    conv = L.Conv2d((H, W, cphys), K, R, stride=stride, pad=pad,
                    groups=groups, relu=True, device=dev, batch=B,
                    needs_dgrad=(Cin > 16))
    x = torch.randn(B, H, W, cphys, device=dev).to(torch.bfloat16)
    if Cin > 16:
        x.requires_grad_()
    return conv, x


def median(f, reps=3):
    ts = sorted(f() for _ in range(reps))
    return ts[reps // 2]


def timed(fn, iters):
    for _ in range(2):
        fn()
    torch.cuda.synchronize()

    def one():
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters
    return median(one)


def force_fwd(conv, M, v):
    conv._v_fwd = v
    conv._v_M = M


def force_dgrad(conv, N, v):
    for p2 in conv.plan.dgrad_plans:
        if p2 is not None:
            p2["variant"] = v
            p2["vN"] = N


def fwd_ok(v, M, Kdf, BNtot):
    if v in (3, 4) and M < 2048:
        return False
    if v == 4 and Kdf < 192:
        return False
    return True


def tune_shape(row, iters, table, report, stages=("fwd", "dgrad",
                                                  "wgrad")):
    name, B, H, W, Cin, K, R, stride, pad, groups = row
    from theanompi_amd.ops import tuning
    conv, x = build(*row)
    plan = conv.plan
    M = B * plan.P * plan.Q
    flops = 2.0 * B * plan.P * plan.Q * K * (Cin // groups) * R * R

    # -- fwd sweep ----------------------------------------------------------
    if "fwd" in stages:
        with torch.no_grad():
            xn = x.detach()
            results = {}
            force_fwd(conv, M, 0)
            results[0] = timed(lambda: conv(xn), iters)
            for v in FWD_VARIANTS:
                if not fwd_ok(v, M, plan.Kdf, plan.Kgp):
                    continue
                force_fwd(conv, M, v)
                results[v] = timed(lambda: conv(xn), iters)
        best_f = min(results, key=results.get)
        t0f, tbf = results[0], results[best_f]
        if best_f != 0 and tbf < t0f * 0.99:
            table[tuning.key("fwd", M, plan.Kgp, plan.Kdf, plan.G)] = \
                best_f
        force_fwd(conv, M, best_f if tbf < t0f * 0.99 else 0)
        report.append(
            f"| {name} fwd | {M}x{plan.Kgp}x{plan.Kdf}g{plan.G} "
            f"| v{best_f} | {t0f*1e3:.3f} -> {tbf*1e3:.3f} ms "
            f"| {flops/tbf/1e12:.0f} TF | "
            + " ".join(f"v{v}:{t*1e3:.3f}"
                       for v, t in sorted(results.items())) + " |")
    else:
        force_fwd(conv, M, 0)

    if not x.requires_grad:
        del conv, x
        return

    # -- dgrad sweep (fwd pinned; wgrad at default) --------------------------
    dy = torch.randn(B, plan.P, plan.Q, plan.Kphys, device=x.device) \
        .to(torch.bfloat16)

    def fb():
        y = conv(x)
        y.backward(dy)
        x.grad = None
        conv.weight.grad = None
        if conv.bias is not None:
            conv.bias.grad = None

    if "dgrad" in stages:
        dres = {}
        force_dgrad(conv, B, 0)
        dres[0] = timed(fb, iters)
        for v in DGRAD_VARIANTS:
            force_dgrad(conv, B, v)
            dres[v] = timed(fb, iters)
        best_d = min(dres, key=dres.get)
        if best_d != 0 and dres[best_d] < dres[0] * 0.995:
            for p2 in plan.dgrad_plans:
                if p2 is not None:
                    table[tuning.key("dgrad", B * p2["Hi"] * p2["Wj"],
                                     plan.Cg, p2["Kdf2"], plan.G)] = \
                        best_d
        force_dgrad(conv, B,
                    best_d if dres[best_d] < dres[0] * 0.995 else 0)
        report.append(
            f"| {name} dgrad | (all classes) | v{best_d} "
            f"| {dres[0]*1e3:.3f} -> {dres[best_d]*1e3:.3f} ms | | "
            + " ".join(f"v{v}:{t*1e3:.3f}"
                       for v, t in sorted(dres.items())) + " |")

    # -- wgrad sweep ---------------------------------------------------------
    if "wgrad" in stages:
        wres = {}
        conv._v_wgrad = (0, 0)
        wres[(0, 0)] = timed(fb, iters)
        for vt in WGRAD_GRID:
            conv._v_wgrad = vt
            wres[vt] = timed(fb, iters)
        best_w = min(wres, key=wres.get)
        if best_w != (0, 0) and wres[best_w] < wres[(0, 0)] * 0.995:
            table[tuning.key("wgrad", M, plan.Kgp, plan.Kdf, plan.G)] = \
                list(best_w)
        report.append(
            f"| {name} wgrad | {M}x{plan.Kgp}x{plan.Kdf}g{plan.G} "
            f"| v{best_w} | {wres[(0,0)]*1e3:.3f} -> "
            f"{wres[best_w]*1e3:.3f} ms | | "
            + " ".join(f"{v}/{t_}:{t*1e3:.3f}"
                       for (v, t_), t in sorted(wres.items())) + " |")
    del conv, x, dy
    torch.cuda.empty_cache()


def model_rows(name):
    """Instantiate a zoo model at its bench batch size and enumerate
    EVERY conv/linear layer's exact GEMM shape (deduped) — full
    coverage instead of hand-picked representatives."""
    import importlib
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    from bench import MODELS  # repo-root bench.py
    from theanompi_amd.models import layers as L
    modname, clsname, batch = MODELS[name]
    cls = getattr(importlib.import_module(modname), clsname)
    model = cls({"rank": 0, "size": 1, "device": "cuda:0",
                 "batch_size": batch, "n_iters": 2})
    rows, seen = [], set()
    for lname, mod in model.module.named_modules():
        if not isinstance(mod, (L.Conv2d, L.Linear)):
            continue
        p = getattr(mod, "plan", None)
        if p is None:
            continue
        cin = getattr(mod, "Cg_logical", p.Cg) * p.G
        k = p.Kg * p.G
        key = (p.NHWC, p.R, p.stride, p.pad, p.G, k)
        if key in seen:
            continue
        seen.add(key)
        rows.append((f"{name}.{lname}", p.NHWC[0], p.NHWC[1], p.NHWC[2],
                     cin, k, p.R, p.stride[0], p.pad[0], p.G))
    del model
    torch.cuda.empty_cache()
    return rows


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="all", choices=["all"] + sorted(SHAPES))
    ap.add_argument("--models", default="",
                    help="comma list of zoo models: tune every layer "
                         "shape of each (overrides --model)")
    ap.add_argument("--iters", type=int, default=30)
    ap.add_argument("--stages", default="fwd,dgrad,wgrad",
                    help="comma subset of fwd,dgrad,wgrad")
    ap.add_argument("--out", default="gpurun_out/conv_tuned.json")
    ap.add_argument("--report", default="gpurun_out/autotune_report.md")
    ap.add_argument("--base", default="",
                    help="existing table to merge new winners into")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    os.environ.setdefault("THEANOMPI_TUNED_OFF", "1")  # sweep from clean

    table = {}
    if args.base:
        with open(args.base) as f:
            table = json.load(f)
    report = ["| stage | gemm | winner | default -> best | TF | all |",
              "|---|---|---|---|---|---|"]
    stages = tuple(args.stages.split(","))
    if args.models:
        rows = []
        for name in args.models.split(","):
            rows.extend(model_rows(name.strip()))
        for row in rows:
            tune_shape(row, args.iters, table, report, stages)
            print(report[-1], flush=True)
        keys = []
    else:
        keys = sorted(SHAPES) if args.model == "all" else [args.model]
    for k in keys:
        for row in SHAPES[k]:
            tune_shape(row, args.iters, table, report, stages)
            print(report[-1], flush=True)

    os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
    with open(args.out, "w") as f:
        json.dump(table, f, indent=1, sort_keys=True)
    with open(args.report, "w") as f:
        f.write("# conv autotune sweep (MI355X)\n\n")
        f.write("\n".join(report) + "\n")
    print(f"wrote {len(table)} tuned entries -> {args.out}")


if __name__ == "__main__":
    main()
