"""Python-stack samples -> pystacks.csv (unified schema).

Parity: reference bin/sofa_preprocess.py:1709-1761 (pairs of timestamp/stack
lines; durations by diffing consecutive timestamps; idle frames dropped) —
extended to pyflame's whole-process view: the sampler emits every thread, the
header line carries `<ts> <tid> <thread-name>` and durations are diffed PER
THREAD.  Legacy 1-token headers (main thread only) still parse.
Input: pystacks.txt.<pid> files written by pystacks_inject/sitecustomize.py.
"""

from __future__ import annotations

import glob
import os
from collections import defaultdict
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
        # per-thread sample streams
        by_tid = defaultdict(lambda: ([], [], []))  # ts, stacks, names
        for i in range(0, len(lines) - 1, 2):
            head = lines[i].split()
            if not head:
                continue
            try:
                t = float(head[0])
            except ValueError:
                continue
            tid = int(head[1]) if len(head) > 1 else 0
            tname = head[2] if len(head) > 2 else ""
            ts, stacks, tnames = by_tid[tid]
            ts.append(t)
            stacks.append(lines[i + 1])
            tnames.append(tname)
        for tid, (ts, stacks, tnames) in by_tid.items():
            if not ts:
                continue
            t_arr = np.array(ts)
            dur = np.diff(
                t_arr,
                append=t_arr[-1] + (t_arr[-1] - t_arr[0]) / max(len(t_arr) - 1, 1),
            )
            keep = [
                not any(s.split(";")[-1].startswith(f) for f in IDLE_FRAMES)
                for s in stacks
            ]
            n = int(np.sum(keep))
            if n == 0:
                continue
            df = new_trace_df(n)
            sel_ts = t_arr[keep]
            df["timestamp"] = (sel_ts - tb.time_base) if tb is not None else sel_ts
            df["duration"] = dur[keep]
            df["pid"] = pid
            df["tid"] = tid
            df["name"] = [
                (("[%s] " % nm) if nm else "") + s.replace(";", "<br>")
                for s, nm, k in zip(stacks, tnames, keep)
                if k
            ]
            df["category"] = 3
            frames.append(df)
    if not frames:
        return new_trace_df(0)
    out = pd.concat(frames, ignore_index=True)
    out.sort_values("timestamp", inplace=True, kind="stable")
    out.reset_index(drop=True, inplace=True)
    return out
