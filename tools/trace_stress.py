#!/usr/bin/env python3
"""Collector trace-rate ceiling: launch-storm stress.

Fires tiny kernels in a tight multi-stream loop under the collector
(kernel-dispatch + filtered HIP API tracing) and reports the sustained
launch rate, collector events/s, and drop count — the "events/sec at
high kernel-launch rates" half of the BASELINE metric, measured at its
ceiling rather than at a workload's natural rate.

Run on a GPU box: python tools/trace_stress.py [--launches 200000]
"""

import argparse
import ctypes
import glob
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
TRACER = os.path.join(REPO, "sofa_amd", "native", "lib", "libsofatracer.so")


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--launches", type=int, default=200_000)
    ap.add_argument("--streams", type=int, default=4)
    ap.add_argument("--tracer", choices=["sdk", "lite"], default="lite")
    args = ap.parse_args()

    logdir = os.path.join(REPO, "gpurun_out", "stress_sgt")
    os.makedirs(logdir, exist_ok=True)
    from bench import setup_tracer_env, TracerCtl

    setup_tracer_env(logdir, args.tracer)
    if args.tracer == "sdk":
        os.environ["SOFA_TRACE_HIP_API"] = "1"

    import torch

    assert torch.cuda.is_available()
    lib = TracerCtl(args.tracer)

    streams = [torch.cuda.Stream() for _ in range(args.streams)]
    x = [torch.zeros(64, device="cuda") for _ in range(args.streams)]

    def storm(n):
        per = n // args.streams
        for s, xs in zip(streams, x):
            with torch.cuda.stream(s):
                for _ in range(per):
                    xs.add_(1.0)  # one tiny kernel per call
        torch.cuda.synchronize()

    # plain rate
    storm(2000)
    t0 = time.perf_counter()
    storm(args.launches)
    plain_s = time.perf_counter() - t0
    plain_rate = args.launches / plain_s

    # traced rate
    lib.start()
    storm(2000)
    n0 = lib.event_count()
    t0 = time.perf_counter()
    storm(args.launches)
    traced_s = time.perf_counter() - t0
    lib.stop()
    n_events = int(lib.event_count() - n0)
    traced_rate = args.launches / traced_s

    from sofa_amd.preprocess.sgt import parse_sgt

    dropped = 0
    recorded = 0
    for p in glob.glob(os.path.join(logdir, "gputrace_*.sgt")):
        s = parse_sgt(p)
        dropped += s.dropped
        recorded += s.n_events

    print("launch storm: %d launches over %d streams" % (args.launches, args.streams))
    print("plain:  %8.0f launches/s" % plain_rate)
    print("traced: %8.0f launches/s  (%.1f%% overhead)"
          % (traced_rate, 100 * (plain_s and (traced_s - plain_s) / plain_s)))
    print("collector: %d events in %.3f s = %8.0f events/s captured; dropped %d"
          % (n_events, traced_s, n_events / traced_s, dropped))
    print("trace file events total: %d" % recorded)
    return 0


if __name__ == "__main__":
    sys.exit(main())
