"""blktrace.txt (tracefs block tracepoints) -> per-IO latency rows.

Parity with the reference's blkparse pipeline: match block_rq_issue to
block_rq_complete by (device, sector) to compute per-request latency
(cyliustack/sofa bin/sofa_preprocess.py:684-781 — including NOT repeating
its `'D' is event` string-identity bug).  Output rows use the unified
schema: timestamp = issue time, duration = latency (s), payload = bytes,
bandwidth = bytes/latency, deviceId = major:minor packed, event = sector.

Line shapes handled (kernel version dependent; io-prio field optional):
  task-123 [002] d..2. 6447.914364: block_rq_issue: 8,0 W 524288 () 2048 + 1024 none,0 [dd]
  <idle>-0 [002] d.h2. 6447.920001: block_rq_complete: 8,0 W () 2048 + 1024 none,0 [0]
"""

from __future__ import annotations

import os
import re
from collections import defaultdict, deque
from typing import Optional

import numpy as np

from ..schema import new_trace_df, trace_df_from
from .timebase import TimeBase

LINE_RE = re.compile(
    r"^\s*(?P<task>.+?)-(?P<pid>\d+)\s+\[(?P<cpu>\d+)\]\s+\S+\s+"
    r"(?P<ts>[\d.]+):\s+block_rq_(?P<kind>issue|complete):\s+"
    r"(?P<maj>\d+),(?P<min>\d+)\s+(?P<rwbs>\S+)"
    r"(?:\s+(?P<bytes>\d+))?\s+\([^)]*\)\s+"
    r"(?P<sector>\d+)\s+\+\s+(?P<nr>\d+)"
)


def parse_blkio(logdir: str, tb: Optional[TimeBase]):
    """Returns a unified-schema DataFrame of completed IOs (empty if no file)."""
    path = os.path.join(logdir, "blktrace.txt")
    if not os.path.isfile(path):
        return new_trace_df(0)

    # (dev, sector) -> FIFO of (issue_ts_s, bytes, rwbs, pid, task)
    pending = defaultdict(deque)
    rows = []  # (t_issue, latency, bytes, dev_packed, sector, rwbs, pid, task)
    with open(path, errors="replace") as f:
        for line in f:
            m = LINE_RE.match(line)
            if m is None:
                continue
            ts = float(m.group("ts"))
            dev = (int(m.group("maj")) << 20) | int(m.group("min"))
            sector = int(m.group("sector"))
            nr = int(m.group("nr"))
            if m.group("kind") == "issue":
                nbytes = int(m.group("bytes") or 0) or nr * 512
                pending[(dev, sector)].append(
                    (ts, nbytes, m.group("rwbs"), int(m.group("pid")), m.group("task").strip())
                )
            else:
                q = pending.get((dev, sector))
                if not q:
                    continue
                t0, nbytes, rwbs, pid, task = q.popleft()
                lat = ts - t0
                if lat < 0:
                    continue
                rows.append((t0, lat, nbytes, dev, sector, rwbs, pid, task))

    if not rows:
        return new_trace_df(0)
    t0 = np.array([r[0] for r in rows])
    lat = np.array([r[1] for r in rows])
    nbytes = np.array([r[2] for r in rows], dtype=np.int64)
    dev = np.array([r[3] for r in rows], dtype=np.int64)
    sector = np.array([r[4] for r in rows], dtype=np.float64)
    # trace_clock is mono_raw (record/blkio.py) -> one subtraction to timeline
    if tb is not None:
        ts_timeline = tb.raw_to_timeline((t0 * 1e9).astype(np.int64))
    else:
        ts_timeline = t0
    with np.errstate(divide="ignore", invalid="ignore"):
        bw = np.where(lat > 0, nbytes / np.maximum(lat, 1e-12), 0.0)
    names = np.array(
        [
            "block:%s %d,%d sector=%d %d bytes lat=%.3f ms [%s]"
            % (r[5], r[3] >> 20, r[3] & 0xFFFFF, r[4], r[2], r[1] * 1e3, r[7])
            for r in rows
        ],
        dtype=object,
    )
    return trace_df_from(
        len(rows),
        timestamp=ts_timeline,
        duration=lat,
        payload=nbytes,
        bandwidth=bw,
        deviceId=dev,
        event=sector,
        pid=np.array([r[6] for r in rows], dtype=np.int64),
        name=names,
        category=np.full(len(rows), 6, dtype=np.int64),  # 6 = block-IO
    )
