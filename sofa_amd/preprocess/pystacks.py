"""Python-stack samples -> pystacks.csv (unified schema).

Parity: reference bin/sofa_preprocess.py:1709-1761 (pairs of timestamp/stack
lines; durations by diffing consecutive timestamps; idle frames dropped).
Input: pystacks.txt.<pid> files written by pystacks_inject/sitecustomize.py.
"""

from __future__ import annotations

import glob
import os
from typing import Optional

import numpy as np
import pandas as pd

from ..schema import new_trace_df
from .timebase import TimeBase

IDLE_FRAMES = ("wait (", "select (", "poll (", "_sampler (", "sleep (")


def parse_pystacks(logdir: str, tb: Optional[TimeBase]) -> pd.DataFrame:
    frames = []
    for path in sorted(glob.glob(os.path.join(logdir, "pystacks.txt.*"))):
        pid = int(path.rsplit(".", 1)[1])
        try:
            with open(path) as f:
                lines = f.read().splitlines()
        except OSError:
            continue
        ts, stacks = [], []
        for i in range(0, len(lines) - 1, 2):
            try:
                t = float(lines[i])
            except ValueError:
                continue
            ts.append(t)
            stacks.append(lines[i + 1])
        if not ts:
            continue
        t_arr = np.array(ts)
        dur = np.diff(t_arr, append=t_arr[-1] + (t_arr[-1] - t_arr[0]) / max(len(t_arr) - 1, 1))
        keep = [not any(s.split(";")[-1].startswith(f) for f in IDLE_FRAMES) for s in stacks]
        df = new_trace_df(int(np.sum(keep)))
        sel_ts = t_arr[keep]
        df["timestamp"] = (sel_ts - tb.time_base) if tb is not None else sel_ts
        df["duration"] = dur[keep]
        df["pid"] = pid
        df["name"] = [s.replace(";", "<br>") for s, k in zip(stacks, keep) if k]
        df["category"] = 3
        frames.append(df)
    if not frames:
        return new_trace_df(0)
    return pd.concat(frames, ignore_index=True)
