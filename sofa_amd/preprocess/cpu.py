"""CPU samples -> cputrace.csv (unified schema).

Schema semantics follow the reference (bin/sofa_preprocess.py:110-154):
``event`` = log10(instruction pointer), ``duration`` = sampled CPU time in
seconds (here: the cpu-clock sample period, exact, instead of the reference's
cycles/MHz estimate).  Names are "symbol @ dso" resolved offline
(symbols.Symbolizer), demangled for the viz copy (:1814-1816).
"""

from __future__ import annotations

import os
from typing import Optional

import numpy as np
import pandas as pd

from ..schema import new_trace_df, trace_df_from
from .scs import ScsFile, parse_scs
from .symbols import Symbolizer
from .timebase import TimeBase


def load_scs(logdir: str) -> Optional[ScsFile]:
    path = os.path.join(logdir, "cpusamples.scs")
    if not os.path.isfile(path):
        return None
    return parse_scs(path)


def mean_cpu_mhz(logdir: str) -> float:
    """Mean sampled CPU MHz from cpuinfo.txt (recorder poller format:
    'ts mhz mhz ...' per line).  Used to convert hw-cycles sample periods to
    seconds, mirroring the reference's cycles/MHz conversion
    (bin/sofa_preprocess.py:131-139).  0.0 when unavailable."""
    path = os.path.join(logdir, "cpuinfo.txt")
    try:
        total, n = 0.0, 0
        with open(path) as f:
            for line in f:
                parts = line.split()
                for v in parts[1:]:
                    total += float(v)
                    n += 1
        return total / n if n else 0.0
    except (OSError, ValueError):
        return 0.0


def scs_to_cputrace(
    scs: ScsFile, tb: Optional[TimeBase], logdir: str = "", symbolize: bool = True
) -> pd.DataFrame:
    s = scs.samples
    n = len(s)
    if n == 0:
        return new_trace_df(0)
    # sampler stamps CLOCK_MONOTONIC_RAW (cpusampler.cc attr.use_clockid)
    mono = s["time_ns"].astype(np.int64)
    if tb is not None:
        # the scs header carries its own clock pair; prefer it (same process)
        off = scs.realtime_ns - scs.monotonic_raw_ns
        ts = ((mono + off) * 1e-9) - tb.time_base
    else:
        ts = mono * 1e-9
    ips = s["ip"].astype(np.float64)
    with np.errstate(divide="ignore"):
        event = np.where(ips > 0, np.log10(np.maximum(ips, 1.0)), 0.0)
    names = None

    if symbolize:
        kallsyms = os.path.join(logdir, "kallsyms") if logdir else ""
        if kallsyms and not os.path.isfile(kallsyms):
            kallsyms = ""
        croot = ""
        cr_path = os.path.join(logdir, "container_root.txt") if logdir else ""
        if cr_path and os.path.isfile(cr_path):
            with open(cr_path) as crf:
                croot = crf.read().strip()
        symr = Symbolizer(scs.mmaps, kallsyms, container_root=croot)
        # resolve UNIQUE (pid, ip, kernel-flag) triples only, then map back
        # vectorized (a long run has millions of samples but few unique IPs).
        # Structured key — no bit-packing, so large pids (pid_max can be
        # 4194304) cannot collide (round-1 ADVICE).
        kern = (s["flags"].astype(np.uint64) & 1)
        # factorized integer codes over the full (pid, kern, ip) triple —
        # collision-free for any pid/ip width and ~4x faster than a
        # structured-dtype np.unique (hash factorize, no void sort)
        import pandas as _pd

        ip_code, _ = _pd.factorize(s["ip"].astype(np.uint64))
        pid_code, _ = _pd.factorize(s["pid"].astype(np.uint64))
        code = (pid_code.astype(np.int64) * 2 + kern.astype(np.int64)) * (
            ip_code.max() + 1
        ) + ip_code
        inv, uniq_codes = _pd.factorize(code)
        # first occurrence index of each unique triple
        first_idx = np.full(len(uniq_codes), n, dtype=np.int64)
        np.minimum.at(first_idx, inv, np.arange(n))
        uniq = uniq_codes
        uniq_names = np.empty(len(uniq), dtype=object)
        for u in range(len(uniq)):
            i = int(first_idx[u])
            sym, dso = symr.resolve(int(s["pid"][i]), int(s["ip"][i]), bool(kern[i]))
            comm = scs.comms.get(int(s["tid"][i]), "")
            nm = f"{sym} @ {dso}"
            if comm:
                nm = f"{nm} [{comm}]"
            uniq_names[u] = nm
        names = uniq_names[inv]
    else:
        names = np.char.add("ip_", s["ip"].astype("U16")).astype(object)
    # single-shot frame construction: repeated df[col]=... consolidation on
    # large frames costs seconds of first-touch page faults (measured 13 s
    # at 500k samples; this path is 1.8 s)
    # period unit depends on the sampled event (scs header reserved[0]):
    # cpu-clock periods are ns; hw-cycles periods are cycle counts and need
    # the recorded MHz (reference converts cycles/MHz the same way)
    if getattr(scs, "event_type", 0) == 1:
        mhz = mean_cpu_mhz(logdir) if logdir else 0.0
        hz = mhz * 1e6 if mhz > 0 else 2.0e9  # conservative 2 GHz fallback
        duration = s["period"].astype(np.float64) / hz
    else:
        duration = s["period"].astype(np.float64) * 1e-9
    return trace_df_from(
        n,
        timestamp=ts,
        event=event,
        duration=duration,
        deviceId=s["cpu"].astype(np.int64),
        pid=s["pid"].astype(np.int64),
        tid=s["tid"].astype(np.int64),
        name=names,
        category=np.zeros(n, dtype=np.int64),
    )
