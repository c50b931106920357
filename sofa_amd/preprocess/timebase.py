"""Clock bases for the unified timeline.

Every collector stamps (CLOCK_REALTIME, CLOCK_MONOTONIC_RAW) pairs itself
(cpusampler header; SGT header + REC_CLOCK records; timebase.json), so
placing a stream on the timeline is one subtraction — unlike the reference's
perf-uptime/epoch/GPU-clock pairing heuristics (bin/sofa_preprocess.py:
1553-1616,1765-1784).  Timeline coordinate = seconds since `sofa_time.txt`
(epoch at record start), matching the reference's convention.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass
from typing import Optional


@dataclass
class TimeBase:
    time_base: float            # epoch seconds at record start
    realtime_ns: int            # one correlation instant ...
    monotonic_raw_ns: int       # ... in both clocks

    def raw_to_timeline(self, mono_raw_ns):
        """CLOCK_MONOTONIC_RAW ns -> seconds since record start (vectorized)."""
        offset_ns = self.realtime_ns - self.monotonic_raw_ns
        return ((mono_raw_ns + offset_ns) * 1e-9) - self.time_base

    def epoch_to_timeline(self, epoch_s):
        return epoch_s - self.time_base


def load_timebase(logdir: str, cpu_time_offset_ms: int = 0) -> Optional[TimeBase]:
    tb_path = os.path.join(logdir, "timebase.json")
    st_path = os.path.join(logdir, "sofa_time.txt")
    if not os.path.isfile(st_path):
        return None
    with open(st_path) as f:
        time_base = float(f.read().strip()) + cpu_time_offset_ms * 1e-3
    realtime_ns = monotonic_raw_ns = None
    if os.path.isfile(tb_path):
        with open(tb_path) as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                d = json.loads(line)
                realtime_ns = d["realtime_ns"]
                monotonic_raw_ns = d["monotonic_raw_ns"]
                break
    if realtime_ns is None:
        # degenerate: treat monotonic_raw as epoch-aligned (tests/synthetic)
        realtime_ns = int(time_base * 1e9)
        monotonic_raw_ns = 0
    return TimeBase(time_base=time_base, realtime_ns=realtime_ns, monotonic_raw_ns=monotonic_raw_ns)
