#!/usr/bin/env python3
"""Overhead decomposition: where does GPU-tracer overhead come from?

Runs the BASELINE workload (ResNet-50 bs=64 bf16, synthetic) in SEPARATE
processes — one per collector configuration — so every config is measured
against a truly-plain baseline (in-process A/B understates lite-mode cost
because intercept queues exist from init).  Reference methodology:
cyliustack/sofa validation/framework_eval.py:50-99 (same workload with and
without the profiler, repeated).

Configs:
  plain       no tool libraries
  lite-off    hsalite loaded, queues proxied, nothing instrumented (proxy floor)
  lite-prof   + hsa_amd_profiling_set_profiler_enabled (CP timestamping cost)
  lite-full   + signal attach/reap = the shipped lite collector
  sdk-null    rocprofiler-sdk collector, records discarded (SDK floor)
  sdk-full    rocprofiler-sdk collector (round-1 default)

Usage: python tools/overhead_decomp.py [--steps 30] [--warmup 10] [--reps 3]
Writes gpurun_out/overhead_decomp.json and prints a table.
"""

import argparse
import json
import os
import statistics
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(REPO, "sofa_amd", "native", "lib")

CHILD = r"""
import os, sys, time
sys.path.insert(0, %(repo)r)
import torch
from sofa_amd.workloads.resnet import build_resnet50
import torch.nn as nn
torch.manual_seed(0)
dev = "cuda:0"
m = build_resnet50(device=dev, channels_last=True)
opt = torch.optim.SGD(m.parameters(), lr=0.1, momentum=0.9)
x = torch.randn(%(batch)d, 3, 224, 224, device=dev).to(memory_format=torch.channels_last)
t = torch.randint(0, 1000, (%(batch)d,), device=dev)
lf = nn.CrossEntropyLoss()
def step():
    opt.zero_grad(set_to_none=True)
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        loss = lf(m(x), t)
    loss.backward(); opt.step()
for _ in range(%(warmup)d):
    step()
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(%(steps)d):
    step()
torch.cuda.synchronize()
t1 = time.perf_counter()
print("MS_PER_STEP %%.4f" %% ((t1 - t0) / %(steps)d * 1e3))
"""


def configs(logdir):
    lite = os.path.join(LIB, "libsofahsalite.so")
    sdk = os.path.join(LIB, "libsofatracer.so")
    base = {"SOFA_LOGDIR": logdir}
    return [
        ("plain", {}),
        ("lite-off", {**base, "HSA_TOOLS_LIB": lite, "SOFA_LITE_MODE": "off"}),
        ("lite-prof", {**base, "HSA_TOOLS_LIB": lite, "SOFA_LITE_MODE": "prof"}),
        ("lite-full", {**base, "HSA_TOOLS_LIB": lite}),
        ("lite-rs", {**base, "HSA_TOOLS_LIB": lite, "SOFA_LITE_REPLACE_SIGNALS": "1"}),
        ("lite-nospans", {**base, "HSA_TOOLS_LIB": lite, "SOFA_LITE_SUBMIT_SPANS": "0"}),
        ("lite-sample4", {**base, "HSA_TOOLS_LIB": lite, "SOFA_LITE_SAMPLE": "4"}),
        ("lite-sample16", {**base, "HSA_TOOLS_LIB": lite, "SOFA_LITE_SAMPLE": "16"}),
        ("sdk-null", {**base, "ROCP_TOOL_LIBRARIES": sdk, "SOFA_NULL_SINK": "1",
                      "SOFA_TRACE_HIP_API": "1", "SOFA_TRACE_RCCL": "1"}),
        ("sdk-full", {**base, "ROCP_TOOL_LIBRARIES": sdk,
                      "SOFA_TRACE_HIP_API": "1", "SOFA_TRACE_RCCL": "1"}),
    ]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--reps", type=int, default=3)
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--only", default="", help="comma-separated config names")
    args = ap.parse_args()

    outdir = os.path.join(REPO, "gpurun_out")
    logdir = os.path.join(outdir, "decomp_sgt")
    os.makedirs(logdir, exist_ok=True)
    code = CHILD % {
        "repo": REPO,
        "batch": args.batch,
        "steps": args.steps,
        "warmup": args.warmup,
    }
    only = set(args.only.split(",")) if args.only else None
    results = {}
    for name, extra in configs(logdir):
        if only and name not in only:
            continue
        times = []
        for rep in range(args.reps):
            env = dict(os.environ)
            env.update(extra)
            r = subprocess.run(
                [sys.executable, "-c", code],
                env=env,
                capture_output=True,
                text=True,
                timeout=900,
            )
            ms = None
            for line in r.stdout.splitlines():
                if line.startswith("MS_PER_STEP"):
                    ms = float(line.split()[1])
            if ms is None:
                print(f"[{name} rep{rep}] FAILED:\n{r.stdout[-1500:]}\n{r.stderr[-1500:]}")
                continue
            times.append(ms)
            print(f"[{name} rep{rep}] {ms:.3f} ms/step", flush=True)
        if times:
            results[name] = {
                "ms_per_step": times,
                "median": statistics.median(times),
                "min": min(times),
            }

    if "plain" in results:
        base = results["plain"]["median"]
        print(f"\n{'config':<12} {'median ms':>10} {'min ms':>10} {'overhead%':>10}")
        for name, r in results.items():
            ov = 100.0 * (r["median"] - base) / base
            print(f"{name:<12} {r['median']:>10.3f} {r['min']:>10.3f} {ov:>10.2f}")
    with open(os.path.join(outdir, "overhead_decomp.json"), "w") as f:
        json.dump(results, f, indent=1)
    print("\nwrote gpurun_out/overhead_decomp.json")


if __name__ == "__main__":
    main()
