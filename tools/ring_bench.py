#!/usr/bin/env python3
"""Throughput benchmark for the on-device trace-ring compaction kernel
(gfx950).  Prints records/s and effective GB/s for a range of ring sizes.

Run on a GPU box:  python tools/ring_bench.py
"""

import ctypes
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(REPO, "sofa_amd", "native", "lib", "libsofahip.so")

REC_BYTES = 32


def main():
    lib = ctypes.CDLL(LIB)
    lib.sofa_ring_create.argtypes = [ctypes.c_int, ctypes.c_uint32, ctypes.POINTER(ctypes.c_void_p)]
    lib.sofa_ring_destroy.argtypes = [ctypes.c_void_p]
    lib.sofa_ring_test_produce.argtypes = [ctypes.c_void_p, ctypes.c_uint32, ctypes.c_uint32]
    lib.sofa_ring_compact.argtypes = [
        ctypes.c_void_p, ctypes.c_uint64, ctypes.c_double, ctypes.c_longlong,
        ctypes.c_void_p, ctypes.c_uint32, ctypes.POINTER(ctypes.c_uint32),
    ]
    lib.sofa_ring_compact_bench.argtypes = [
        ctypes.c_void_p, ctypes.c_uint64, ctypes.c_int, ctypes.POINTER(ctypes.c_double)
    ]

    print("n_records,produce_s,kernel_ms,records_per_s,effective_GBps,e2e_s,kept")
    import os
    sizes = (20, 22, 24, 25, 27) if os.environ.get("RING_BENCH_BIG") else (20, 22, 24, 25)
    for log2n in sizes:
        n = 1 << log2n
        ring = ctypes.c_void_p()
        assert lib.sofa_ring_create(0, n, ctypes.byref(ring)) == 0
        t0 = time.perf_counter()
        assert lib.sofa_ring_test_produce(ring, n, 7) == 0
        t1 = time.perf_counter()
        # kernel-only throughput (device buffers, hipEvent timing)
        ms = ctypes.c_double(0)
        assert lib.sofa_ring_compact_bench(ring, (1 << 2) | (1 << 4), 20, ctypes.byref(ms)) == 0
        recs_per_s = n / (ms.value * 1e-3)
        # traffic: read 32B/record + write 32B/kept (kept = 2/7 of records)
        gbps = n * REC_BYTES * (1 + 2.0 / 7.0) / (ms.value * 1e-3) / 1e9
        # end-to-end incl. alloc + D2H of kept records
        host = np.zeros(n * REC_BYTES, dtype=np.uint8)
        n_out = ctypes.c_uint32(0)
        t2 = time.perf_counter()
        assert lib.sofa_ring_compact(
            ring, (1 << 2) | (1 << 4), 10.0, 12345,
            host.ctypes.data_as(ctypes.c_void_p), n, ctypes.byref(n_out)) == 0
        t3 = time.perf_counter()
        print("%d,%.4f,%.4f,%.3e,%.1f,%.4f,%d"
              % (n, t1 - t0, ms.value, recs_per_s, gbps, t3 - t2, n_out.value))
        lib.sofa_ring_destroy(ring)


if __name__ == "__main__":
    sys.exit(main())
