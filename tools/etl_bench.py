#!/usr/bin/env python3
"""ETL throughput benchmark: SGT binary -> unified-schema DataFrame events/s.

The reference's preprocess collapses at modern kernel-launch rates
(nvprof-CSV + per-row Python); this measures the rebuilt
vectorized path (numpy run-detection + columnar assembly).  Runs anywhere
(synthetic trace, no GPU).

Usage: python tools/etl_bench.py [--n 1000000]
"""

import argparse
import os
import sys
import tempfile
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=1_000_000)
    args = ap.parse_args()

    from sgt_synth import SgtWriter
    from sofa_amd.preprocess.gpu import sgt_to_gputrace, sgt_to_rccltrace
    from sofa_amd.preprocess.sgt import KERNEL_DTYPE, COPY_DTYPE, parse_sgt
    from sofa_amd.preprocess.timebase import TimeBase

    n = args.n
    n_kernels = int(n * 0.9)
    n_copies = n - n_kernels
    w = SgtWriter(pid=1, realtime_ns=10**18, rocp_ns=0)
    w.clock(realtime_ns=10**18, mono=0, rocp=0)
    w.agent(handle=1, device=0)
    for kid in range(200):
        w.kernel_name(kid, f"_Z14fused_kernel_{kid}IfEvPT_S1_mm")

    # bulk-generate records as numpy (fast writer for the fixture)
    ks = np.zeros(n_kernels, KERNEL_DTYPE)
    ks["type"] = 1
    ks["size"] = KERNEL_DTYPE.itemsize
    t = np.arange(n_kernels, dtype=np.uint64) * 2000
    ks["start_ns"] = t
    ks["end_ns"] = t + 1500
    ks["kernel_id"] = np.arange(n_kernels) % 200
    ks["device"] = np.arange(n_kernels) % 8
    ks["tid"] = 7
    cs = np.zeros(n_copies, COPY_DTYPE)
    cs["type"] = 2
    cs["size"] = COPY_DTYPE.itemsize
    t2 = np.arange(n_copies, dtype=np.uint64) * 20000
    cs["start_ns"] = t2
    cs["end_ns"] = t2 + 5000
    cs["op"] = 2
    cs["src_device"] = -1
    cs["dst_device"] = 0
    cs["bytes"] = 1 << 16
    w.buf += ks.tobytes() + cs.tobytes()

    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "gputrace_1.sgt")
        w.write(path)
        size_mb = os.path.getsize(path) / 1e6

        # warmup (numpy/pandas/demangle cold-start excluded from the measure)
        tb = TimeBase(time_base=10**9, realtime_ns=10**18, monotonic_raw_ns=0)
        sgt_to_gputrace([parse_sgt(path)], tb)

        t0 = time.perf_counter()
        sgt = parse_sgt(path)
        t1 = time.perf_counter()
        df = sgt_to_gputrace([sgt], tb)
        t2p = time.perf_counter()

        n_ev = sgt.n_events
        print(f"trace: {size_mb:.1f} MB, {n_ev} events")
        print(f"parse (binary->arrays):      {t1 - t0:.3f} s = {n_ev / (t1 - t0):,.0f} events/s")
        print(f"to unified schema (pandas):  {t2p - t1:.3f} s = {n_ev / (t2p - t1):,.0f} events/s")
        print(f"end-to-end ETL:              {t2p - t0:.3f} s = {n_ev / (t2p - t0):,.0f} events/s")
        assert len(df) == n_ev


if __name__ == "__main__":
    main()
