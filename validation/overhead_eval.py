#!/usr/bin/env python3
"""Overhead-validation harness — the reference's headline methodology.

Parity with cyliustack/sofa validation/framework_eval.py:
  * collect mode (:50-99): run each benchmark N times WITH and WITHOUT
    `sofa record`, store per-run step times;
  * report mode (:188-215): keep the fastest half of each population,
    report mean/std, a paired t-test p-value, and
    `mean of overheads (%)` = 100 * mean(|with - without| / without).

Benchmarks here are the repo's own workloads (ResNet-50 / Llama step via
bench.py's plain phase), synthetic data, random init.

Usage:
  python validation/overhead_eval.py collect --num-runs 10 [--steps 20] \
      [--out validation/overhead_runs.json]
  python validation/overhead_eval.py report [--in validation/overhead_runs.json]
"""

import argparse
import json
import os
import subprocess
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(REPO, "bench.py")


def run_once(steps: int, warmup: int, profiled: bool) -> dict:
    args = [sys.executable, BENCH, "--steps", str(steps), "--warmup", str(warmup)]
    if not profiled:
        args.append("--no-profile")
    r = subprocess.run(args, capture_output=True, text=True, timeout=1200, cwd=REPO)
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    if not lines:
        raise RuntimeError(r.stderr[-2000:])
    d = json.loads(lines[-1])
    return {
        "ms_per_step": d["ms_per_step"],
        "ms_per_step_plain": d["config"]["ms_per_step_plain"],
        "events_per_sec": d["value"],
    }


def collect(args) -> None:
    runs = {"with": [], "without": []}
    for i in range(args.num_runs):
        print(f"run {i + 1}/{args.num_runs} (with profiler)")
        runs["with"].append(run_once(args.steps, args.warmup, True)["ms_per_step"])
        print(f"run {i + 1}/{args.num_runs} (without profiler)")
        runs["without"].append(run_once(args.steps, args.warmup, False)["ms_per_step"])
    with open(args.out, "w") as f:
        json.dump(runs, f, indent=1)
    print(f"wrote {args.out}")


def report(args) -> None:
    from scipy import stats

    with open(getattr(args, "in")) as f:
        runs = json.load(f)
    w = np.sort(np.array(runs["with"], dtype=float))
    wo = np.sort(np.array(runs["without"], dtype=float))
    # fastest half (reference :200-206)
    half = max(len(w) // 2, 1)
    w, wo = w[:half], wo[:half]
    overhead = 100.0 * np.mean(np.abs(w - wo) / wo)
    t, pval = stats.ttest_rel(w, wo) if len(w) > 1 else (0.0, 1.0)
    print("with-profiler    ms/step: mean %.3f std %.3f" % (w.mean(), w.std()))
    print("without-profiler ms/step: mean %.3f std %.3f" % (wo.mean(), wo.std()))
    print("paired t-test p-value: %.4f" % pval)
    print("mean of overheads (%%): %.2f" % overhead)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("verb", choices=["collect", "report"])
    ap.add_argument("--num-runs", type=int, default=10)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--out", default=os.path.join(REPO, "validation", "overhead_runs.json"))
    ap.add_argument("--in", dest="in", default=os.path.join(REPO, "validation", "overhead_runs.json"))
    args = ap.parse_args()
    (collect if args.verb == "collect" else report)(args)


if __name__ == "__main__":
    main()
