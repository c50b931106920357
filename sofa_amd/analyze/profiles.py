"""Per-subsystem profiles, each appending (name, value) features.

Parity map (reference bin/sofa_analyze.py):
  cpu_profile :694-710, gpu_profile :343-377, nvsmi_profile :259-341,
  mpstat_profile :735-790, vmstat_profile :712-733, diskstat_profile
  :640-692, netbandwidth_profile :531-594, net_profile :385-493,
  spotlight ROI :873-896.
"""

from __future__ import annotations

import os
from typing import List, Tuple

import numpy as np
import pandas as pd

from .. import printing as p

Features = List[Tuple[str, float]]


def _q(series) -> Tuple[float, float, float, float]:
    a = np.asarray(series, dtype=np.float64)
    if len(a) == 0:
        return (0.0, 0.0, 0.0, 0.0)
    return tuple(np.percentile(a, [25, 50, 75, 100]))


def cpu_profile(df_cpu: pd.DataFrame, features: Features, elapsed: float) -> None:
    if df_cpu is None or len(df_cpu) == 0:
        return
    total = df_cpu["duration"].sum()
    features.append(("cpu_time", float(total)))
    features.append(("n_cpu_samples", float(len(df_cpu))))
    print("\nCPU profile:")
    print("  sampled cpu time: %.3f s over %d samples" % (total, len(df_cpu)))
    top = (
        df_cpu.groupby("name")["duration"]
        .agg(["sum", "count"])
        .sort_values("sum", ascending=False)
        .head(10)
    )
    print("  top functions:")
    for name, row in top.iterrows():
        short = name if len(str(name)) < 90 else str(name)[:87] + "..."
        print("    %8.3f s %6d  %s" % (row["sum"], int(row["count"]), short))


def gpu_profile(df_gpu: pd.DataFrame, df_rccl: pd.DataFrame, features: Features) -> None:
    if df_gpu is None or len(df_gpu) == 0:
        return
    print("\nGPU profile:")
    kernels = df_gpu[df_gpu["copyKind"] == 0]
    copies = df_gpu[df_gpu["copyKind"] != 0]
    gpu_time = df_gpu["duration"].sum()
    features.append(("gpu_time", float(gpu_time)))
    features.append(("gpu_kernel_time", float(kernels["duration"].sum())))
    features.append(("gpu_memcpy_time", float(copies["duration"].sum())))
    features.append(("n_gpu_events", float(len(df_gpu))))
    # rccl time: prefer the API trace, fall back to kernel-name grep
    # (reference greps `nccl` in kernel names, bin/sofa_analyze.py:363-368)
    if df_rccl is not None and len(df_rccl):
        rccl_time = df_rccl["duration"].sum()
    else:
        names = kernels["name"].astype(str)
        rccl_time = kernels[names.str.contains("rccl|nccl|Ccl", case=False, regex=True)][
            "duration"
        ].sum()
    features.append(("rccl_kernel_time", float(rccl_time)))
    for dev, grp in df_gpu.groupby("deviceId"):
        print(
            "  gpu%-2d  time %8.4f s  kernels %6d  copies %6d"
            % (
                dev,
                grp["duration"].sum(),
                (grp["copyKind"] == 0).sum(),
                (grp["copyKind"] != 0).sum(),
            )
        )
    top = (
        kernels.groupby("name")["duration"]
        .agg(["sum", "count"])
        .sort_values("sum", ascending=False)
        .head(10)
    )
    if len(top):
        print("  top kernels:")
        for name, row in top.iterrows():
            short = str(name) if len(str(name)) < 90 else str(name)[:87] + "..."
            print("    %8.4f s %6d  %s" % (row["sum"], int(row["count"]), short))

    # achieved kernel concurrency per device: sum(kernel durations) over the
    # union of their busy intervals.  1.0 = fully serialized streams; >1
    # means overlapping streams are actually overlapping on-device.  Pairs
    # with the launch-gap analysis (launch.py) to separate "launch-bound"
    # from "stream-serialized".
    for dev, grp in kernels.groupby("deviceId"):
        if len(grp) < 2:
            continue
        t0 = grp["timestamp"].to_numpy(dtype=np.float64)
        t1 = t0 + grp["duration"].to_numpy(dtype=np.float64)
        order = np.argsort(t0, kind="stable")
        t0, t1 = t0[order], t1[order]
        # merged-interval union, vectorized: a new busy segment starts where
        # this kernel begins after the running max of previous ends
        run_end = np.maximum.accumulate(t1)
        gap_starts = t0[1:] > run_end[:-1]
        seg_start = np.concatenate(([t0[0]], t0[1:][gap_starts]))
        seg_end = np.concatenate((run_end[:-1][gap_starts], [run_end[-1]]))
        union = float(np.maximum(seg_end - seg_start, 0).sum())
        total = float(grp["duration"].sum())
        if union > 0:
            cf = total / union
            features.append((f"gpu{int(dev)}_concurrency_factor", float(cf)))
            print(
                "  gpu%-2d kernel concurrency: %.2fx (busy %.4f s, serialized "
                "sum %.4f s)" % (dev, cf, union, total)
            )

    # chip-underfill: MI355X = 256 CUs across 8 XCDs; launches with fewer
    # than 256 workgroups cannot fill the chip (payload carries the WG count
    # for kernel rows — preprocess.gpu).  Report the GPU-time share spent in
    # such launches; a high share means small-kernel/launch-shape problems
    # that no amount of per-kernel tuning fixes.
    kt = kernels["duration"].sum()
    if kt > 0 and (kernels["payload"] > 0).any():
        under = kernels[(kernels["payload"] > 0) & (kernels["payload"] < 256)]
        ratio = float(under["duration"].sum() / kt)
        features.append(("gpu_underfill_time_ratio", ratio))
        if ratio > 0.01:
            print(
                "  chip underfill: %.1f%% of kernel time in launches with <256 "
                "workgroups (256 CUs / 8 XCDs need >=256 WGs to fill)" % (ratio * 100)
            )
            worst = (
                under.groupby("name")["duration"].sum().sort_values(ascending=False).head(3)
            )
            for name, dur in worst.items():
                short = str(name) if len(str(name)) < 80 else str(name)[:77] + "..."
                print("    %8.4f s  %s" % (dur, short))


def gpusmi_profile(df_sm: pd.DataFrame, features: Features, logdir: str = "") -> None:
    if df_sm is None or len(df_sm) == 0:
        return
    print("\nGPU utilization (rocm-smi):")
    for dev, grp in df_sm.groupby("deviceId"):
        q25, q50, q75, q100 = _q(grp["duration"])
        print("  gpu%-2d busy%%: q25=%.0f q50=%.0f q75=%.0f max=%.0f" % (dev, q25, q50, q75, q100))
        features.append((f"gpu{dev}_util_q50", float(q50)))
        features.append((f"gpu{dev}_util_max", float(q100)))
    # VRAM/power from the telemetry CSV (wider than the trace series)
    path = os.path.join(logdir, "gpusmi_trace.csv") if logdir else ""
    if path and os.path.isfile(path):
        try:
            d = pd.read_csv(path)
            if "vram_MB" in d:
                for dev, grp in d.groupby("dev"):
                    features.append((f"gpu{dev}_vram_peak_MB", float(grp["vram_MB"].max())))
                    if "power_W" in grp:
                        features.append((f"gpu{dev}_power_mean_W", float(grp["power_W"].mean())))
                print(
                    "  VRAM peak %.0f MB; mean power %.0f W (per-GPU detail in features.csv)"
                    % (d["vram_MB"].max(), d.get("power_W", pd.Series([0])).mean())
                )
        except (OSError, ValueError, KeyError):
            pass


def mpstat_profile(df_mp: pd.DataFrame, features: Features, idle_threshold: float = 10.0) -> None:
    if df_mp is None or len(df_mp) == 0:
        return
    print("\nCPU cores (mpstat):")
    busy_by_core = df_mp.groupby("deviceId")["duration"].mean()
    active = (busy_by_core > idle_threshold).sum()
    features.append(("cpu_active_ratio", float(active) / max(len(busy_by_core), 1)))
    features.append(("cpu_mean_busy", float(busy_by_core.mean())))
    print(
        "  %d/%d cores active (>%.0f%% busy); mean busy %.1f%%"
        % (active, len(busy_by_core), idle_threshold, busy_by_core.mean())
    )


def vmstat_profile(logdir: str, features: Features) -> None:
    path = os.path.join(logdir, "vmstat.csv")
    if not os.path.isfile(path):
        return
    try:
        d = pd.read_csv(path)
    except (OSError, ValueError):
        return
    if len(d) == 0:
        return
    print("\nVM stats:")
    for col, feat in [("ctxt_r", "ctxt_per_s"), ("pgpgin_r", "pgin_per_s"), ("pgpgout_r", "pgout_per_s")]:
        if col in d:
            features.append((feat, float(d[col].mean())))
    print(
        "  ctx-switch/s mean %.0f; pgin/s %.0f; pgout/s %.0f"
        % (d.get("ctxt_r", pd.Series([0])).mean(), d.get("pgpgin_r", pd.Series([0])).mean(), d.get("pgpgout_r", pd.Series([0])).mean())
    )


def diskstat_profile(logdir: str, features: Features) -> None:
    path = os.path.join(logdir, "diskstat_vector.csv")
    if not os.path.isfile(path):
        return
    try:
        d = pd.read_csv(path)
    except (OSError, ValueError):
        return
    if len(d) == 0:
        return
    print("\nDisk profile:")
    for dev, grp in d.groupby("dev"):
        rb = grp["read_Bps"].mean() / 1e6
        wb = grp["write_Bps"].mean() / 1e6
        if rb + wb < 0.01:
            continue
        print(
            "  %-10s read %8.1f MB/s  write %8.1f MB/s  r_await %6.2f ms  w_await %6.2f ms"
            % (dev, rb, wb, grp["r_await_ms"].mean(), grp["w_await_ms"].mean())
        )
    features.append(("disk_read_Bps", float(d["read_Bps"].mean())))
    features.append(("disk_write_Bps", float(d["write_Bps"].mean())))


def pc_hotspot_profile(logdir: str, features: Features) -> None:
    """GPU instruction-level hotspots from PC samples (pcsamples.csv,
    --pc_sampling): per-kernel sample share, SIMD-lane activity (divergence
    signal, /64), and the hottest code-object offsets.  Beyond-reference
    capability."""
    path = os.path.join(logdir, "pcsamples.csv")
    if not os.path.isfile(path):
        return
    try:
        d = pd.read_csv(path)
    except (OSError, pd.errors.ParserError):
        return
    if len(d) == 0:
        return
    total = len(d)
    features.append(("pcsamples_total", float(total)))
    features.append(("pcsamples_mean_active_lanes", float(d["active_lanes"].mean())))
    print("\nGPU PC-sample hotspots (%d samples):" % total)
    print("%-58s %8s %7s %9s" % ("kernel", "samples", "share", "lanes/64"))
    g = d.groupby("kernel").agg(
        samples=("offset", "size"), lanes=("active_lanes", "mean")
    ).sort_values("samples", ascending=False)
    for name, row in g.head(8).iterrows():
        short = str(name) if len(str(name)) < 58 else str(name)[:55] + "..."
        print("%-58s %8d %6.1f%% %9.1f" % (
            short, row["samples"], 100.0 * row["samples"] / total, row["lanes"]))
    top_kernel = g.index[0]
    hot = d[d["kernel"] == top_kernel]
    offs = hot.groupby("offset").size().sort_values(ascending=False).head(3)
    print("  hottest offsets in %s:" % (str(top_kernel)[:60]))
    for off, cnt in offs.items():
        print("    +0x%x  %d samples (%.1f%% of kernel)" % (
            int(off), cnt, 100.0 * cnt / len(hot)))


def xgmi_measured_profile(logdir: str, features: Features) -> None:
    """Measured per-xGMI-link bandwidth from gpu_metrics HW accumulators
    (xgmi_counters.csv) — the ground truth the analytic ring model in
    comm.rccl_link_attribution estimates; report both so a model/HW mismatch
    is visible (each MI355X link peaks ~153 GB/s)."""
    path = os.path.join(logdir, "xgmi_counters.csv")
    if not os.path.isfile(path):
        return
    try:
        d = pd.read_csv(path)
    except (OSError, pd.errors.ParserError):
        return
    if len(d) == 0:
        return
    print("\nMeasured xGMI link bandwidth (gpu_metrics accumulators):")
    print("%-6s %-5s %-6s %10s %10s" % ("gpu", "link", "dir", "mean GB/s", "max GB/s"))
    for (dev, link, kind), grp in d.groupby(["dev", "link", "kind"]):
        print("%-6d %-5d %-6s %10.2f %10.2f"
              % (dev, link, kind, grp["GBps"].mean(), grp["GBps"].max()))
    features.append(("xgmi_meas_max_GBps", float(d["GBps"].max())))
    features.append(("xgmi_meas_links_active", float(d.groupby(["dev", "link"]).ngroups)))


def blkio_latency_profile(df_blk: pd.DataFrame, features: Features) -> None:
    """Per-IO latency summary per device (reference blktrace_latency_profile,
    bin/sofa_analyze.py:596-638 — btt replaced by our matched issue/complete
    rows from preprocess.blkio)."""
    if df_blk is None or len(df_blk) == 0:
        return
    print("\nBlock-IO latency profile (per request):")
    print("%-10s %8s %12s %10s %10s %10s %12s" % (
        "device", "IOs", "bytes(MB)", "q25(ms)", "q50(ms)", "q95(ms)", "MB/s"))
    for dev, grp in df_blk.groupby("deviceId"):
        maj, minr = int(dev) >> 20, int(dev) & 0xFFFFF
        lat = grp["duration"]
        q25, q50, _, _ = _q(lat)
        q95 = float(lat.quantile(0.95))
        total_b = grp["payload"].sum()
        total_t = lat.sum()
        mbps = total_b / max(total_t, 1e-9) / 1e6
        print("%-10s %8d %12.2f %10.3f %10.3f %10.3f %12.1f" % (
            f"{maj},{minr}", len(grp), total_b / 1e6,
            q25 * 1e3, q50 * 1e3, q95 * 1e3, mbps))
    features.append(("blkio_num_requests", float(len(df_blk))))
    features.append(("blkio_latency_q50", float(df_blk["duration"].quantile(0.5))))
    features.append(("blkio_latency_q95", float(df_blk["duration"].quantile(0.95))))
    features.append(("blkio_total_bytes", float(df_blk["payload"].sum())))


def netbandwidth_profile(logdir: str, features: Features) -> None:
    path = os.path.join(logdir, "netbandwidth.csv")
    if not os.path.isfile(path):
        return
    try:
        d = pd.read_csv(path)
    except (OSError, ValueError):
        return
    if len(d) == 0:
        return
    print("\nNetwork bandwidth:")
    for col, feat in [("rx_Bps", "net_rx"), ("tx_Bps", "net_tx")]:
        q25, q50, q75, q100 = _q(d[col])
        print("  %s: q25=%.2f q50=%.2f q75=%.2f max=%.2f MB/s" % (feat, q25 / 1e6, q50 / 1e6, q75 / 1e6, q100 / 1e6))
        features.append((feat + "_q50", float(q50)))
        features.append((feat + "_max", float(q100)))


def net_profile(logdir: str, df_net: pd.DataFrame, features: Features) -> None:
    """Packet src->dst ranking -> netrank.csv (reference :385-493)."""
    if df_net is None or len(df_net) == 0:
        return
    g = (
        df_net.groupby(["pkt_src", "pkt_dst"])
        .agg(packets=("payload", "count"), bytes=("payload", "sum"))
        .reset_index()
        .sort_values("bytes", ascending=False)
    )
    g.to_csv(os.path.join(logdir, "netrank.csv"), index=False)
    features.append(("net_num_peers", float(len(g))))
    print("\nNetwork peers (top 5 by bytes):")
    print(g.head(5).to_string(index=False))


def spotlight_roi(df_sm: pd.DataFrame, up: float = 50.0, down: float = 10.0, trigger: int = 10):
    """Hysteresis ROI over GPU busy% (reference :873-896).

    Counts up while busy >= `up`, down while busy < `down`; the first window
    whose count reaches `trigger` opens the ROI, the last closes it.
    """
    if df_sm is None or len(df_sm) == 0:
        return (0.0, 0.0)
    d = df_sm.sort_values("timestamp")
    count = 0
    begin = end = 0.0
    for ts, util in zip(d["timestamp"], d["duration"]):
        if util >= up:
            count += 1
            if count >= trigger and begin == 0.0:
                begin = ts
            if begin > 0.0:
                end = ts
        elif util < down:
            count = max(0, count - 1)
    return (begin, end)
