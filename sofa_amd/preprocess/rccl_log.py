"""RCCL debug-log channel -> rccltrace.csv rows (lite-mode collective args).

In lite mode there is no rocprofiler-sdk in the process, so collective
arguments come from RCCL's own logging: the recorder sets
NCCL_DEBUG=INFO + NCCL_DEBUG_SUBSYS=COLL + NCCL_DEBUG_FILE=<logdir>/
rccl_debug.%h.%p and this parser reads the per-process logs (SURVEY.md §2.7
"RCCL_KERNEL_COLL_TRACE" channel).  One line per collective call:

  [ts] host:pid:tid [dev] NCCL INFO AllReduce: opCount 0 sendbuff 0x...
      recvbuff 0x... count 1048576 datatype 7 op 0 root 0 comm 0x...
      [nranks=8] stream 0x... task 0 globalrank 0

The log carries no fine-grained timestamps (RCCL's optional prefix is
second-resolution), so rows get timestamp 0 and are ORDER-matched to their
ncclDevKernel spans per (pid, device) by analyze.comm.attach_kernel_times —
the same join used for SDK records.
"""

from __future__ import annotations

import glob
import os
import re

import numpy as np

from ..schema import new_trace_df, trace_df_from

LINE_RE = re.compile(
    r"(?P<host>\S+):(?P<pid>\d+):(?P<tid>\d+)\s+\[(?P<dev>\d+)\]\s+NCCL INFO\s+"
    r"(?P<coll>AllReduce|Broadcast|Reduce|AllGather|ReduceScatter|AllToAllv?|"
    r"Send|Recv|Gather|Scatter):\s+opCount\s+(?P<op>[0-9a-fx]+)\s+.*?"
    r"count\s+(?P<count>\d+)\s+datatype\s+(?P<dt>\d+)\s+op\s+(?P<redop>\d+)\s+"
    r"root\s+(?P<root>-?\d+)\s+comm\s+(?P<comm>0x[0-9a-f]+)\s+"
    r"\[nranks=(?P<nranks>\d+)\]\s+stream\s+(?P<stream>0x[0-9a-f]+)"
)

# ncclDataType_t element sizes (same table as the collector)
ELEM_SIZE = (1, 1, 4, 4, 8, 8, 2, 4, 8, 2, 1, 1)


def parse_rccl_log(logdir: str):
    """All rccl_debug.* files -> unified rows (copyKind 16, timestamp 0)."""
    rows = []
    for path in sorted(glob.glob(os.path.join(logdir, "rccl_debug.*"))):
        try:
            with open(path, errors="replace") as f:
                for order, line in enumerate(f):
                    m = LINE_RE.search(line)
                    if m is None:
                        continue
                    count = int(m.group("count"))
                    dt = int(m.group("dt"))
                    elem = ELEM_SIZE[dt] if 0 <= dt < len(ELEM_SIZE) else 1
                    rows.append(
                        (
                            int(m.group("pid")),
                            int(m.group("tid")),
                            int(m.group("dev")),
                            "nccl" + m.group("coll"),
                            count,
                            dt,
                            count * elem,
                            int(m.group("root")),
                            int(m.group("comm"), 16),
                            int(m.group("stream"), 16),
                            order,
                        )
                    )
        except OSError:
            continue
    if not rows:
        return new_trace_df(0)
    n = len(rows)
    payload = np.array([r[6] for r in rows], dtype=np.int64)
    names = np.array(
        [
            "%s(count=%d, dtype=%d, comm=%x, stream=%x)" % (r[3], r[4], r[5], r[8], r[9])
            for r in rows
        ],
        dtype=object,
    )
    return trace_df_from(
        n,
        # timestamp 0 + monotone order in `event`: analyze order-matches to
        # kernel spans; nothing here lands on the wall-clock timeline itself
        timestamp=np.array([r[10] * 1e-9 for r in rows]),
        duration=np.zeros(n),
        deviceId=np.array([r[2] for r in rows], dtype=np.int64),
        copyKind=np.full(n, 16, dtype=np.int64),
        payload=payload,
        pkt_src=np.array([r[8] & 0x7FFFFFFF for r in rows], dtype=np.int64),
        pkt_dst=np.array([r[7] for r in rows], dtype=np.int64),
        pid=np.array([r[0] for r in rows], dtype=np.int64),
        tid=np.array([r[1] for r in rows], dtype=np.int64),
        event=np.array([r[10] for r in rows], dtype=np.float64),
        name=names,
    )
