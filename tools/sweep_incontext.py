#!/usr/bin/env python3
"""IN-CONTEXT tuned-table sweep: A/B candidate conv_tuned.json overlays
through the REAL end-to-end bench instead of isolated microbenches.

Round-2 measured (twice) that per-shape microbench winners can REGRESS
a whole model (GoogLeNet fwd/dgrad entries): isolated kernel timing
misses the in-net L2/launch/DVFS context.  This harness closes that
gap — each candidate is one overlay table; candidates run INTERLEAVED
(round-robin) so box thermal drift hits every candidate equally, and
the report compares medians plus per-position values.

    python tools/sweep_incontext.py --model alexnet \
        --key "wgrad:93312x128x1216g2" --values "[3,4096]" "[4,4096]" \
        --reps 4 --steps 15 [--out gpurun_out/incontext.md]

Multiple --key/--values pairs sweep a cross product capped at
--max-candidates.  Candidate 0 is always the unmodified baseline.
Run under gpurun; each bench invocation is a fresh process so the
tuning table reloads from the overlay (THEANOMPI_TUNED_JSON).
"""
import argparse
import itertools
import json
import os
import statistics
import subprocess
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_bench(model, steps, warmup, env):
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", model, "--steps",
         str(steps), "--warmup", str(warmup), "--no-secondary"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
        env=env)
    for ln in out.stdout.splitlines():
        if ln.startswith("{"):
            return json.loads(ln)["value"]
    raise RuntimeError(
        f"bench failed rc={out.returncode}: {out.stderr[-1500:]}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="alexnet")
    ap.add_argument("--key", action="append", default=[],
                    help="tuned-table key to sweep (repeatable)")
    ap.add_argument("--values", action="append", nargs="+", default=[],
                    help="JSON values for the matching --key")
    ap.add_argument("--reps", type=int, default=4)
    ap.add_argument("--steps", type=int, default=15)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--max-candidates", type=int, default=6)
    ap.add_argument("--out", default="")
    args = ap.parse_args()
    assert len(args.key) == len(args.values), "--key/--values must pair"

    # cross product of per-key value lists -> candidate overlay dicts
    cands = [{}]                               # 0 = baseline, no overlay
    if args.key:
        lists = [[json.loads(v) for v in vs] for vs in args.values]
        for combo in itertools.product(*lists):
            cands.append(dict(zip(args.key, combo)))
    cands = cands[:args.max_candidates]

    tmpdir = tempfile.mkdtemp(prefix="incontext_")
    envs = []
    for i, ov in enumerate(cands):
        env = dict(os.environ)
        if ov:
            path = os.path.join(tmpdir, f"ov{i}.json")
            with open(path, "w") as f:
                json.dump(ov, f)
            env["THEANOMPI_TUNED_JSON"] = path
        envs.append(env)

    results = [[] for _ in cands]
    for rep in range(args.reps):                 # interleaved: c0 c1 ... cN
        for i, env in enumerate(envs):
            v = run_bench(args.model, args.steps, args.warmup, env)
            results[i].append(v)
            print(f"rep {rep} cand {i}: {v:.1f}", flush=True)

    lines = [f"# in-context sweep: {args.model}", ""]
    base_med = statistics.median(results[0])
    for i, (ov, vals) in enumerate(zip(cands, results)):
        med = statistics.median(vals)
        tag = "baseline" if not ov else json.dumps(ov)
        lines.append(f"cand {i} ({tag}): median {med:.1f} "
                     f"({med / base_med * 100 - 100:+.1f}% vs baseline) "
                     f"values {[round(v) for v in vals]}")
    report = "\n".join(lines)
    print(report)
    if args.out:
        with open(args.out, "w") as f:
            f.write(report + "\n")


if __name__ == "__main__":
    main()
