"""Syscall trace (SST binary from sofa-syscalltrace) -> strace.csv.

Replaces the reference's strace text parsing (bin/sofa_preprocess.py:
1623-1704) including its noisy-syscall filter and --strace_min_time cut.
Syscall numbers resolve to names via the system unistd_64.h when present,
else a built-in table of common syscalls.
"""

from __future__ import annotations

import os
import re
import struct
from functools import lru_cache
from typing import Dict, Optional

import numpy as np
import pandas as pd

from ..config import SofaConfig
from ..schema import new_trace_df, trace_df_from
from .timebase import TimeBase

MAGIC = 0x31545353
HEADER_SIZE = 32

SYS_DTYPE = np.dtype(
    [
        ("t_enter_ns", "<u8"),
        ("duration_ns", "<u8"),
        ("tid", "<u4"),
        ("sysno", "<u4"),
        ("ret", "<i8"),
    ]
)

# noisy syscalls the reference filters out (bin/sofa_preprocess.py:1623-1635)
NOISY = {
    "clock_gettime", "gettimeofday", "poll", "ppoll", "epoll_wait",
    "epoll_pwait", "futex", "sched_yield", "nanosleep", "clock_nanosleep",
}

_COMMON = {
    0: "read", 1: "write", 2: "open", 3: "close", 4: "stat", 5: "fstat",
    8: "lseek", 9: "mmap", 10: "mprotect", 11: "munmap", 12: "brk",
    13: "rt_sigaction", 14: "rt_sigprocmask", 16: "ioctl", 17: "pread64",
    18: "pwrite64", 19: "readv", 20: "writev", 21: "access", 22: "pipe",
    23: "select", 24: "sched_yield", 28: "madvise", 32: "dup", 33: "dup2",
    35: "nanosleep", 39: "getpid", 41: "socket", 42: "connect", 43: "accept",
    44: "sendto", 45: "recvfrom", 46: "sendmsg", 47: "recvmsg", 49: "bind",
    50: "listen", 56: "clone", 57: "fork", 59: "execve", 60: "exit",
    61: "wait4", 62: "kill", 72: "fcntl", 74: "fsync", 78: "getdents",
    79: "getcwd", 83: "mkdir", 87: "unlink", 89: "readlink", 96: "gettimeofday",
    97: "getrlimit", 102: "getuid", 158: "arch_prctl", 186: "gettid",
    202: "futex", 228: "clock_gettime", 230: "clock_nanosleep",
    231: "exit_group", 232: "epoll_wait", 233: "epoll_ctl", 257: "openat",
    262: "newfstatat", 270: "pselect6", 271: "ppoll", 281: "epoll_pwait",
    202 + 0: "futex",
}


@lru_cache(maxsize=1)
def syscall_names() -> Dict[int, str]:
    out = dict(_COMMON)
    path = "/usr/include/x86_64-linux-gnu/asm/unistd_64.h"
    if os.path.isfile(path):
        try:
            with open(path) as f:
                for line in f:
                    m = re.match(r"#define\s+__NR_(\w+)\s+(\d+)", line)
                    if m:
                        out[int(m.group(2))] = m.group(1)
        except OSError:
            pass
    return out


def parse_sst(logdir: str, tb: Optional[TimeBase], cfg: SofaConfig) -> pd.DataFrame:
    path = os.path.join(logdir, "strace.sst")
    if not os.path.isfile(path) or os.path.getsize(path) <= HEADER_SIZE:
        return new_trace_df(0)
    with open(path, "rb") as f:
        buf = f.read()
    magic, version, rt, mono, _res = struct.unpack_from("<IIQQQ", buf, 0)
    if magic != MAGIC:
        return new_trace_df(0)
    n = (len(buf) - HEADER_SIZE) // SYS_DTYPE.itemsize
    recs = np.frombuffer(buf, dtype=SYS_DTYPE, count=n, offset=HEADER_SIZE)
    if n == 0:
        return new_trace_df(0)
    names_tbl = syscall_names()
    dur = recs["duration_ns"].astype(np.float64) * 1e-9
    keep = dur >= cfg.strace_min_time
    # drop noisy syscalls
    name_arr = np.array([names_tbl.get(int(s), "sys_%d" % s) for s in recs["sysno"]])
    keep &= ~np.isin(name_arr, list(NOISY))
    recs = recs[keep]
    dur = dur[keep]
    name_arr = name_arr[keep]
    if len(recs) == 0:
        return new_trace_df(0)

    offset_ns = rt - mono
    epoch_s = (recs["t_enter_ns"].astype(np.int64) + offset_ns) * 1e-9
    ts = epoch_s - tb.time_base if tb is not None else epoch_s
    return trace_df_from(
        len(recs),
        timestamp=ts,
        duration=dur,
        tid=recs["tid"].astype(np.int64),
        event=recs["sysno"].astype(np.float64),
        payload=np.maximum(recs["ret"], 0),
        name=np.array(
            [
                "%s(ret=%d) %.1f us" % (nm, r, d * 1e6)
                for nm, r, d in zip(name_arr, recs["ret"], dur)
            ],
            dtype=object,
        ),
        category=np.full(len(recs), 2, dtype=np.int64),
    )
