"""Export the unified timeline as a Chrome/Perfetto trace-event JSON.

Beyond the reference (its only viewer was sofaboard): `chrome_trace.json`
loads in chrome://tracing or ui.perfetto.dev, with one process lane per
stream and one thread lane per device/core.
"""

from __future__ import annotations

import json
import os
from typing import Optional

import pandas as pd


def _events_from_trace(df: pd.DataFrame, pid_label: str, tid_col: str = "deviceId",
                       cat: str = "gpu", max_events: int = 500000):
    out = []
    if df is None or len(df) == 0:
        return out
    if len(df) > max_events:
        df = df.iloc[:: len(df) // max_events + 1]
    ts = (df["timestamp"].to_numpy() * 1e6)  # us
    dur = (df["duration"].to_numpy() * 1e6)
    tids = df[tid_col].to_numpy()
    names = df["name"].astype(str).to_numpy()
    for i in range(len(df)):
        out.append(
            {
                "name": names[i][:120],
                "cat": cat,
                "ph": "X",
                "ts": round(float(ts[i]), 3),
                "dur": max(round(float(dur[i]), 3), 0.001),
                "pid": pid_label,
                "tid": int(tids[i]),
            }
        )
    return out


def write_chrome_trace(logdir: str, pre: Optional[dict] = None, path: str = "chrome_trace.json") -> Optional[str]:
    pre = pre or {}

    def get(name, csv):
        df = pre.get(name)
        if df is not None and len(df):
            return df
        full = os.path.join(logdir, csv)
        if os.path.isfile(full):
            try:
                return pd.read_csv(full)
            except (OSError, ValueError):
                return None
        return None

    events = []
    df_gpu = get("df_gpu", "gputrace.csv")
    if df_gpu is not None:
        kernels = df_gpu[df_gpu["copyKind"] == 0]
        copies = df_gpu[df_gpu["copyKind"] != 0]
        # device-ring software events get their own track group, laned by
        # producer id (the `src` a kernel passed to ring_push, e.g.
        # workgroup index) instead of being mixed into the kernel lanes
        names = kernels["name"].astype(str)
        devring = kernels[names.str.contains("devring:", regex=False)]
        kernels = kernels[~names.str.contains("devring:", regex=False)]
        if len(devring):
            devring = devring.copy()
            # lane = ring_push's producer id (ring_writer maps RingRec.src
            # into tid), mod 64 to bound the track count
            devring["deviceId"] = devring["tid"].astype(int) % 64
            events += _events_from_trace(devring, "devring", "deviceId", "devring")
        events += _events_from_trace(kernels, "GPU kernels", "deviceId", "kernel")
        events += _events_from_trace(copies, "GPU copies", "deviceId", "memcpy")
    df_rccl = get("df_rccl", "rccltrace.csv")
    if df_rccl is not None:
        events += _events_from_trace(df_rccl, "RCCL", "deviceId", "collective")
    df_hip = get("df_hip", "hip_api_trace.csv")
    if df_hip is not None:
        events += _events_from_trace(df_hip, "HIP API", "tid", "api")
    df_cpu = get("df_cpu", "cputrace.csv")
    if df_cpu is not None:
        events += _events_from_trace(df_cpu, "CPU samples", "tid", "sample")
    df_strace = get("df_strace", "strace.csv")
    if df_strace is not None:
        events += _events_from_trace(df_strace, "syscalls", "tid", "syscall")
    if not events:
        return None
    out_path = os.path.join(logdir, path)
    with open(out_path, "w") as f:
        json.dump({"traceEvents": events, "displayTimeUnit": "ms"}, f)
    return out_path
