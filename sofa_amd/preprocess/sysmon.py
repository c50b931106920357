"""Poller TSVs -> per-subsystem CSVs + unified trace rows.

Replaces the reference's per-line Python parsers for mpstat/diskstat/netstat/
vmstat/nvidia-smi (bin/sofa_preprocess.py:482-1183) with vectorized pandas
delta computations over the SysMonitor TSVs (record/pollers.py formats).
"""

from __future__ import annotations

import os
from typing import Optional

import numpy as np
import pandas as pd

from ..schema import new_trace_df
from .timebase import TimeBase


def _read(logdir: str, name: str, columns):
    path = os.path.join(logdir, name)
    if not os.path.isfile(path) or os.path.getsize(path) == 0:
        return None
    try:
        df = pd.read_csv(path, sep=r"\s+", header=None, names=columns, engine="c")
    except (pd.errors.ParserError, ValueError):
        return None
    if len(df) == 0:
        return None
    return df


def _tl(tb: Optional[TimeBase], epoch: pd.Series) -> np.ndarray:
    if tb is None:
        return epoch.to_numpy()
    return epoch.to_numpy() - tb.time_base


# ------------------------------------------------------------------- mpstat


def parse_mpstat(logdir: str, tb: Optional[TimeBase]):
    """Returns (trace_df, mpstat_csv_df, usr_sys_csv_df) or (empty,...)"""
    cols = ["ts", "core", "usr", "nice", "sys", "idle", "iow", "irq", "sirq", "steal"]
    raw = _read(logdir, "mpstat.txt", cols)
    if raw is None:
        return new_trace_df(0), None, None
    raw = raw.sort_values(["core", "ts"], kind="stable")
    g = raw.groupby("core")
    d = raw.copy()
    for c in ["usr", "nice", "sys", "idle", "iow", "irq", "sirq", "steal"]:
        d[c] = g[c].diff()
    d["dt"] = g["ts"].diff()
    d = d.dropna()
    d = d[d["dt"] > 0]
    if len(d) == 0:
        return new_trace_df(0), None, None
    total = d[["usr", "nice", "sys", "idle", "iow", "irq", "sirq", "steal"]].sum(axis=1)
    total = total.replace(0, np.nan)
    for c in ["usr", "sys", "idle", "iow", "irq"]:
        d[c + "_r"] = (d[c] / total * 100.0).fillna(0.0)
    d["busy_r"] = 100.0 - d["idle_r"]

    tdf = new_trace_df(len(d))
    tdf["timestamp"] = _tl(tb, d["ts"])
    tdf["duration"] = d["busy_r"].to_numpy()  # y value = busy %
    tdf["deviceId"] = d["core"].to_numpy(dtype=np.int64)
    tdf["event"] = d["core"].to_numpy(dtype=np.float64)
    tdf["name"] = [
        "mpstat_core%d (usr:%.0f%%|sys:%.0f%%|idl:%.0f%%|iow:%.0f%%|irq:%.0f%%)"
        % (core, u, s, i, w, q)
        for core, u, s, i, w, q in zip(
            d["core"], d["usr_r"], d["sys_r"], d["idle_r"], d["iow_r"], d["irq_r"]
        )
    ]
    tdf["category"] = 0

    mp_csv = d[["ts", "core", "usr_r", "sys_r", "idle_r", "iow_r", "irq_r", "busy_r"]].copy()
    mp_csv["timestamp"] = _tl(tb, d["ts"])
    usr_sys = d[["ts", "core", "usr_r", "sys_r"]].copy()
    usr_sys["timestamp"] = _tl(tb, d["ts"])
    return tdf, mp_csv, usr_sys


# ----------------------------------------------------------------- diskstat


def parse_diskstat(logdir: str, tb: Optional[TimeBase], sector_bytes: int = 512):
    cols = ["ts", "dev", "reads", "rsect", "rms", "writes", "wsect", "wms", "inflight"]
    raw = _read(logdir, "diskstat.txt", cols)
    if raw is None:
        return new_trace_df(0), None
    raw = raw.sort_values(["dev", "ts"], kind="stable")
    g = raw.groupby("dev")
    d = raw.copy()
    for c in ["reads", "rsect", "rms", "writes", "wsect", "wms"]:
        d[c] = g[c].diff()
    d["dt"] = g["ts"].diff()
    d = d.dropna()
    d = d[d["dt"] > 0]
    if len(d) == 0:
        return new_trace_df(0), None
    d["r_iops"] = d["reads"] / d["dt"]
    d["w_iops"] = d["writes"] / d["dt"]
    d["read_Bps"] = d["rsect"] * sector_bytes / d["dt"]
    d["write_Bps"] = d["wsect"] * sector_bytes / d["dt"]
    d["r_await_ms"] = np.where(d["reads"] > 0, d["rms"] / d["reads"].replace(0, np.nan), 0.0)
    d["w_await_ms"] = np.where(d["writes"] > 0, d["wms"] / d["writes"].replace(0, np.nan), 0.0)
    # drop devices with no activity at all (reference drops all-zero devices)
    active_devs = d.groupby("dev")[["read_Bps", "write_Bps"]].sum().sum(axis=1)
    active_devs = set(active_devs[active_devs > 0].index)
    d = d[d["dev"].isin(active_devs)]
    if len(d) == 0:
        return new_trace_df(0), None

    tdf = new_trace_df(len(d))
    tdf["timestamp"] = _tl(tb, d["ts"])
    tdf["duration"] = ((d["read_Bps"] + d["write_Bps"]) / 1e6).to_numpy()  # MB/s
    tdf["payload"] = ((d["read_Bps"] + d["write_Bps"]) * d["dt"]).to_numpy(dtype=np.int64)
    tdf["bandwidth"] = (d["read_Bps"] + d["write_Bps"]).to_numpy()
    tdf["name"] = [
        "diskstat_%s (R:%.1fMB/s W:%.1fMB/s r_iops:%.0f w_iops:%.0f)"
        % (dev, rb / 1e6, wb / 1e6, ri, wi)
        for dev, rb, wb, ri, wi in zip(
            d["dev"], d["read_Bps"], d["write_Bps"], d["r_iops"], d["w_iops"]
        )
    ]
    vec = d[
        ["ts", "dev", "r_iops", "w_iops", "read_Bps", "write_Bps", "r_await_ms", "w_await_ms", "inflight"]
    ].copy()
    vec["timestamp"] = _tl(tb, d["ts"])
    return tdf, vec


# ------------------------------------------------------------------ netstat


def parse_netstat(logdir: str, tb: Optional[TimeBase]):
    cols = ["ts", "iface", "rx_bytes", "rx_pkts", "tx_bytes", "tx_pkts"]
    raw = _read(logdir, "netstat.txt", cols)
    if raw is None:
        return new_trace_df(0), None
    raw = raw[raw["iface"] != "lo"]
    if len(raw) == 0:
        return new_trace_df(0), None
    raw = raw.sort_values(["iface", "ts"], kind="stable")
    g = raw.groupby("iface")
    d = raw.copy()
    for c in ["rx_bytes", "rx_pkts", "tx_bytes", "tx_pkts"]:
        d[c] = g[c].diff()
    d["dt"] = g["ts"].diff()
    d = d.dropna()
    d = d[d["dt"] > 0]
    if len(d) == 0:
        return new_trace_df(0), None
    d["rx_Bps"] = d["rx_bytes"] / d["dt"]
    d["tx_Bps"] = d["tx_bytes"] / d["dt"]

    tdf = new_trace_df(len(d))
    tdf["timestamp"] = _tl(tb, d["ts"])
    tdf["duration"] = ((d["rx_Bps"] + d["tx_Bps"]) / 1e6).to_numpy()  # MB/s
    tdf["bandwidth"] = (d["rx_Bps"] + d["tx_Bps"]).to_numpy()
    tdf["name"] = [
        "netstat_%s (rx:%.2fMB/s tx:%.2fMB/s)" % (i, r / 1e6, t / 1e6)
        for i, r, t in zip(d["iface"], d["rx_Bps"], d["tx_Bps"])
    ]
    bw = d[["ts", "iface", "rx_Bps", "tx_Bps"]].copy()
    bw["timestamp"] = _tl(tb, d["ts"])
    return tdf, bw


# ------------------------------------------------------------------- vmstat


def parse_vmstat(logdir: str, tb: Optional[TimeBase]):
    cols = ["ts", "pgpgin", "pgpgout", "pswpin", "pswpout", "ctxt", "intr", "running", "blocked"]
    raw = _read(logdir, "vmstat.txt", cols)
    if raw is None:
        return new_trace_df(0), None
    d = raw.copy()
    for c in ["pgpgin", "pgpgout", "pswpin", "pswpout", "ctxt", "intr"]:
        d[c] = d[c].diff()
    d["dt"] = d["ts"].diff()
    d = d.dropna()
    d = d[d["dt"] > 0]
    if len(d) == 0:
        return new_trace_df(0), None
    for c in ["pgpgin", "pgpgout", "pswpin", "pswpout", "ctxt", "intr"]:
        d[c + "_r"] = d[c] / d["dt"]

    tdf = new_trace_df(len(d))
    tdf["timestamp"] = _tl(tb, d["ts"])
    tdf["duration"] = d["ctxt_r"].to_numpy()  # context switches/s as y
    tdf["name"] = [
        "vmstat (bi:%.0f bo:%.0f cs:%.0f in:%.0f run:%d blk:%d)"
        % (bi, bo, cs, it, r, b)
        for bi, bo, cs, it, r, b in zip(
            d["pgpgin_r"], d["pgpgout_r"], d["ctxt_r"], d["intr_r"],
            d["running"].astype(int), d["blocked"].astype(int),
        )
    ]
    vm = d[["ts", "pgpgin_r", "pgpgout_r", "pswpin_r", "pswpout_r", "ctxt_r", "intr_r", "running", "blocked"]].copy()
    vm["timestamp"] = _tl(tb, d["ts"])
    return tdf, vm


# ------------------------------------------------------------------- gpusmi


def parse_gpusmi(logdir: str, tb: Optional[TimeBase]):
    """gpusmi.txt -> (sm_trace, mem_trace, gpusmi_csv).

    Mirrors the reference's nvsmi trace semantics (bin/sofa_preprocess.py:
    1013-1089): event 0 = GPU busy %, event 1 = memory busy %, event 2 = MM
    (media/VCN) engine busy % — the dmon enc/dec analog (:1097-1183); rows
    with -1 MM (no gpu_metrics support) drop the MM series.  Older 6-column
    files (no mm) still parse.
    """
    path = os.path.join(logdir, "gpusmi.txt")
    ncols = 0
    if os.path.isfile(path):
        with open(path) as f:
            first = f.readline().split()
            ncols = len(first)
    cols = ["ts", "dev", "busy", "membusy", "vram", "power"]
    if ncols >= 7:
        cols = cols + ["mm"]
    raw = _read(logdir, "gpusmi.txt", cols)
    if raw is None:
        return new_trace_df(0), new_trace_df(0), None
    if "mm" not in raw.columns:
        raw["mm"] = -1
    # ragged/corrupt rows: coerce everything numeric, drop rows missing the
    # required fields, default the optional ones (fuzz-hardened)
    for c in raw.columns:
        raw[c] = pd.to_numeric(raw[c], errors="coerce")
    raw = raw.dropna(subset=["ts", "dev", "busy"])
    raw["membusy"] = raw["membusy"].fillna(-1)
    raw["vram"] = raw["vram"].fillna(-1)
    raw["power"] = raw["power"].fillna(-1.0)
    raw["mm"] = raw["mm"].fillna(-1)
    raw["dev"] = raw["dev"].astype(np.int64)
    d = raw[raw["busy"] >= 0]
    if len(d) == 0:
        return new_trace_df(0), new_trace_df(0), None

    sm = new_trace_df(len(d))
    sm["timestamp"] = _tl(tb, d["ts"])
    sm["event"] = 0.0
    sm["duration"] = d["busy"].to_numpy(dtype=np.float64)
    sm["deviceId"] = d["dev"].to_numpy(dtype=np.int64)
    sm["name"] = [
        "gpu%d_util:%d%%" % (dev, b) for dev, b in zip(d["dev"], d["busy"])
    ]

    mem = new_trace_df(len(d))
    mem["timestamp"] = _tl(tb, d["ts"])
    mem["event"] = 1.0
    mem["duration"] = d["membusy"].clip(lower=0).to_numpy(dtype=np.float64)
    mem["deviceId"] = d["dev"].to_numpy(dtype=np.int64)
    mem["name"] = [
        "gpu%d_mem:%d%%" % (dev, b) for dev, b in zip(d["dev"], d["membusy"])
    ]

    mm_rows = d[d["mm"] >= 0]
    if len(mm_rows):
        mmt = new_trace_df(len(mm_rows))
        mmt["timestamp"] = _tl(tb, mm_rows["ts"])
        mmt["event"] = 2.0
        mmt["duration"] = mm_rows["mm"].to_numpy(dtype=np.float64)
        mmt["deviceId"] = mm_rows["dev"].to_numpy(dtype=np.int64)
        mmt["name"] = [
            "gpu%d_mm:%d%%" % (dev, b) for dev, b in zip(mm_rows["dev"], mm_rows["mm"])
        ]
        mem = pd.concat([mem, mmt], ignore_index=True)

    csv = d.copy()
    csv["timestamp"] = _tl(tb, d["ts"])
    csv["vram_MB"] = d["vram"].clip(lower=0) / 1e6
    csv["power_W"] = d["power"].clip(lower=0)
    return sm, mem, csv


def parse_xgmi_counters(logdir: str, tb: Optional[TimeBase]):
    """xgmi_counters.txt (per-device HW accumulators: ts dev r0..r7 w0..w7,
    KB) -> per-link measured bandwidth rows + xgmi_counters.csv dataframe.

    This is the MEASURED counterpart of analyze.comm's analytic ring model:
    deltas of the gpu_metrics xgmi_read/write_data_acc counters over the poll
    interval give true per-link GB/s regardless of which collective/algorithm
    produced the traffic.
    """
    cols = ["ts", "dev"] + ["r%d" % i for i in range(8)] + ["w%d" % i for i in range(8)]
    raw = _read(logdir, "xgmi_counters.txt", cols)
    if raw is None or len(raw) < 2:
        return new_trace_df(0), None
    for c in raw.columns:
        raw[c] = pd.to_numeric(raw[c], errors="coerce")
    raw = raw.dropna(subset=["ts", "dev"]).fillna(0)
    if len(raw) < 2:
        return new_trace_df(0), None
    raw["dev"] = raw["dev"].astype(np.int64)
    frames = []
    csv_rows = []
    for dev, grp in raw.groupby("dev"):
        grp = grp.sort_values("ts")
        dt = grp["ts"].diff().to_numpy()
        for link in range(8):
            for kind, pref in (("read", "r"), ("write", "w")):
                acc = grp[f"{pref}{link}"].to_numpy(dtype=np.float64)
                delta_kb = np.diff(acc, prepend=acc[0])
                with np.errstate(divide="ignore", invalid="ignore"):
                    gbps = np.where(
                        (dt > 0) & (delta_kb > 0), delta_kb * 1e3 / np.maximum(dt, 1e-9) / 1e9, 0.0
                    )
                sel = gbps > 0.01  # suppress idle-link noise rows
                if not sel.any():
                    continue
                n = int(sel.sum())
                t = new_trace_df(n)
                t["timestamp"] = _tl(tb, grp["ts"][sel])
                t["duration"] = np.maximum(dt[sel], 0)
                t["deviceId"] = int(dev)
                t["event"] = float(link)
                t["bandwidth"] = gbps[sel] * 1e9
                t["payload"] = (delta_kb[sel] * 1e3).astype(np.int64)
                t["pkt_src"] = int(dev)
                t["pkt_dst"] = link
                t["name"] = [
                    "xgmi gpu%d link%d %s %.2f GB/s" % (dev, link, kind, g)
                    for g in gbps[sel]
                ]
                frames.append(t)
                for ts_v, g in zip(grp["ts"][sel], gbps[sel]):
                    csv_rows.append((ts_v, int(dev), link, kind, g))
    if not frames:
        return new_trace_df(0), None
    trace = pd.concat(frames, ignore_index=True).sort_values("timestamp")
    csv = pd.DataFrame(csv_rows, columns=["ts", "dev", "link", "kind", "GBps"])
    return trace.reset_index(drop=True), csv
