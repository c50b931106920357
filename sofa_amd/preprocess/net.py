"""Packet capture -> nettrace.csv (unified schema).

Replaces the reference's tcpdump text pipeline + per-packet mp.Pool parsing
(bin/sofa_preprocess.py:1188-1231).  Keeps the reference's address encoding:
IPv4 packed base-1000 into pkt_src/pkt_dst (bin/sofa_preprocess.py:182-186)
so comm-report/netrank semantics carry over.  The packet "duration" model uses
measured per-interface bandwidth when available instead of the reference's
hardcoded 128 MB/s (a documented reference bug NOT replicated).
"""

from __future__ import annotations

import os
import struct
from typing import Optional

import numpy as np
import pandas as pd

from ..config import SofaConfig
from ..schema import new_trace_df, trace_df_from
from .timebase import TimeBase

MAGIC = 0x31435053
HEADER_SIZE = 32

PKT_DTYPE = np.dtype(
    [
        ("time_ns", "<u8"),
        ("src_ip", "<u4"),
        ("dst_ip", "<u4"),
        ("sport", "<u2"),
        ("dport", "<u2"),
        ("len", "<u4"),
        ("proto", "u1"),
        ("dir", "u1"),
        ("_pad", "<u2"),
    ]
)


def pack_ip_base1000(ip: np.ndarray) -> np.ndarray:
    """a.b.c.d -> a*1e9 + b*1e6 + c*1e3 + d (reference encoding)."""
    a = (ip >> 24) & 0xFF
    b = (ip >> 16) & 0xFF
    c = (ip >> 8) & 0xFF
    d = ip & 0xFF
    return (
        a.astype(np.int64) * 1000000000
        + b.astype(np.int64) * 1000000
        + c.astype(np.int64) * 1000
        + d.astype(np.int64)
    )


def ip_str(ip: int) -> str:
    return "%d.%d.%d.%d" % ((ip >> 24) & 0xFF, (ip >> 16) & 0xFF, (ip >> 8) & 0xFF, ip & 0xFF)


def parse_pktcap(logdir: str, tb: Optional[TimeBase], cfg: SofaConfig) -> pd.DataFrame:
    path = os.path.join(logdir, "pktcap.bin")
    if not os.path.isfile(path) or os.path.getsize(path) <= HEADER_SIZE:
        return new_trace_df(0)
    with open(path, "rb") as f:
        buf = f.read()
    magic, version, rt, mono, _res = struct.unpack_from("<IIQQQ", buf, 0)
    if magic != MAGIC:
        return new_trace_df(0)
    n = (len(buf) - HEADER_SIZE) // PKT_DTYPE.itemsize
    pkts = np.frombuffer(buf, dtype=PKT_DTYPE, count=n, offset=HEADER_SIZE)
    if n == 0:
        return new_trace_df(0)

    offset_ns = rt - mono
    epoch_s = (pkts["time_ns"].astype(np.int64) + offset_ns) * 1e-9
    ts = epoch_s - tb.time_base if tb is not None else epoch_s

    # duration model: payload / measured NIC bandwidth (fallback 1 GB/s)
    bw = 1e9
    nb_path = os.path.join(logdir, "netbandwidth.csv")
    if os.path.isfile(nb_path):
        try:
            nb = pd.read_csv(nb_path)
            peak = (nb["rx_Bps"] + nb["tx_Bps"]).max()
            if peak and peak > 1e6:
                bw = float(peak)
        except (OSError, KeyError, ValueError):
            pass
    proto_name = np.where(pkts["proto"] == 6, "tcp", np.where(pkts["proto"] == 17, "udp", "ip"))
    names = np.array(
        [
            "network:%s:%s:%d_to_%s:%d_with_%d" % (pr, ip_str(s), sp, ip_str(d), dp, ln)
            for pr, s, sp, d, dp, ln in zip(
                proto_name, pkts["src_ip"], pkts["sport"], pkts["dst_ip"], pkts["dport"], pkts["len"]
            )
        ],
        dtype=object,
    )
    return trace_df_from(
        n,
        timestamp=ts,
        payload=pkts["len"].astype(np.int64),
        duration=pkts["len"].astype(np.float64) / bw,
        bandwidth=np.full(n, bw),
        pkt_src=pack_ip_base1000(pkts["src_ip"]),
        pkt_dst=pack_ip_base1000(pkts["dst_ip"]),
        name=names,
        category=np.zeros(n, dtype=np.int64),
    )
