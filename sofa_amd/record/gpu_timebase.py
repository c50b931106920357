"""GPU timebase prologue: run the CDNA4 s_memrealtime microkernel at record
start and persist the (host CLOCK_MONOTONIC_RAW <-> device tick) correlation
+ measured tick rate to gpu_timebase.json.

This is the product use of native/hip/timebase_kernel.hip (BASELINE.json
north star): preprocess/analyze use it to VALIDATE rocprofiler's
host-correlated timestamps (drift check in analyze).
"""

from __future__ import annotations

import ctypes
import json
import os
from typing import Optional


def _lib_path() -> str:
    return os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "native", "lib", "libsofahip.so",
    )


def sample_gpu_timebase(device: int = 0, rounds: int = 5) -> Optional[dict]:
    """Best-of-N correlation rounds + tick-rate measurement; None if no GPU."""
    path = _lib_path()
    if not os.path.exists(path):
        return None
    try:
        lib = ctypes.CDLL(path)
    except OSError:
        return None
    lib.sofa_gpu_timebase_sample.argtypes = [ctypes.c_int] + [
        ctypes.POINTER(ctypes.c_uint64)
    ] * 4
    lib.sofa_gpu_timebase_freq.argtypes = [
        ctypes.c_int, ctypes.c_int, ctypes.POINTER(ctypes.c_double)
    ]

    best = None
    for _ in range(rounds):
        hb = ctypes.c_uint64(0)
        dmin = ctypes.c_uint64(0)
        dmax = ctypes.c_uint64(0)
        ha = ctypes.c_uint64(0)
        rc = lib.sofa_gpu_timebase_sample(
            device, ctypes.byref(hb), ctypes.byref(dmin), ctypes.byref(dmax), ctypes.byref(ha)
        )
        if rc != 0:
            return None
        window = ha.value - hb.value
        if best is None or window < best["host_window_ns"]:
            best = {
                "host_before_ns": hb.value,
                "host_after_ns": ha.value,
                "host_window_ns": window,
                "device_ticks": dmin.value,
                "device_wave_spread_ticks": dmax.value - dmin.value,
            }
    freq = ctypes.c_double(0)
    if lib.sofa_gpu_timebase_freq(device, 100, ctypes.byref(freq)) == 0:
        best["ticks_per_second"] = freq.value
    else:
        best["ticks_per_second"] = 1e8  # CDNA s_memrealtime nominal 100 MHz
    best["device"] = device

    # MFMA-timed marker: a matrix-core burst that measures its own duration
    # with s_memrealtime.  Any tracer profiling this process records a span
    # for `mfma_marker_kernel`; analyze cross-checks that span against
    # mfma_marker_self_ns (clock-scale validation against on-device ground
    # truth; BASELINE.json north star "MFMA-timed markers").
    try:
        lib.sofa_gpu_mfma_marker.argtypes = [
            ctypes.c_int, ctypes.c_int,
            ctypes.POINTER(ctypes.c_uint64), ctypes.POINTER(ctypes.c_uint64),
            ctypes.POINTER(ctypes.c_uint64), ctypes.POINTER(ctypes.c_double),
        ]
        hb = ctypes.c_uint64(0)
        ha = ctypes.c_uint64(0)
        ticks = ctypes.c_uint64(0)
        self_ns = ctypes.c_double(0)
        iters = 20000  # ~4 MFMA chains x 20k -> O(100 us) marker
        if lib.sofa_gpu_mfma_marker(
            device, iters, ctypes.byref(hb), ctypes.byref(ha),
            ctypes.byref(ticks), ctypes.byref(self_ns),
        ) == 0 and ticks.value > 0:
            best["mfma_marker_iters"] = iters
            best["mfma_marker_self_ticks"] = ticks.value
            best["mfma_marker_self_ns"] = self_ns.value
            best["mfma_marker_host_window_ns"] = ha.value - hb.value
    except (OSError, AttributeError):
        pass
    return best


def write_gpu_timebase(logdir: str, device: int = 0) -> bool:
    info = sample_gpu_timebase(device)
    if info is None:
        return False
    with open(os.path.join(logdir, "gpu_timebase.json"), "w") as f:
        json.dump(info, f, indent=1)
    return True
