"""sofa preprocess — raw logdir -> unified CSVs + report.js.

Orchestration parity with reference bin/sofa_preprocess.py:377-2104, rebuilt
on binary collectors + vectorized parsers.  Every stream is optional: a
missing raw file only warns (the reference's tested no-GPU degradation path,
SURVEY.md §4).
"""

from __future__ import annotations

import os
from typing import List

import numpy as np
import pandas as pd

from .. import printing as p
from ..config import SofaConfig
from ..schema import SOFATrace, new_trace_df, traces_to_json, write_trace_csv
from . import cpu as cpu_mod
from . import gpu as gpu_mod
from . import sysmon
from .timebase import load_timebase

FILTER_COLORS = [
    "red", "orange", "yellow", "green", "blue", "indigo", "violet",
    "cyan", "magenta", "brown",
]


def _filter_traces(df: pd.DataFrame, filters, prefix: str) -> List[SOFATrace]:
    """Per-keyword colored sub-series (reference gpu/cpu_filters behavior,
    bin/sofa_preprocess.py:1439-1452,1818-1827)."""
    out = []
    if df is None or len(df) == 0:
        return out
    names = df["name"].astype(str)
    for i, f in enumerate(filters or []):
        sel = df[names.str.contains(f.keyword, case=False, regex=False)]
        if len(sel) == 0:
            continue
        out.append(
            SOFATrace(
                name=f"{prefix}_filter_{i}",
                title=f"{prefix}:{f.keyword}",
                color=f.color or FILTER_COLORS[i % len(FILTER_COLORS)],
                data=sel,
            )
        )
    return out


def sofa_preprocess(cfg: SofaConfig) -> dict:
    logdir = cfg.logdir
    if not os.path.isdir(logdir):
        p.print_error(f"logdir {logdir} does not exist — run `sofa record` first")
        return {}
    p.print_progress(f"preprocessing {logdir}")
    t_start = __import__("time").perf_counter()
    tb = load_timebase(logdir, cfg.cpu_time_offset_ms)
    traces: List[SOFATrace] = []
    result = {"tb": tb}

    # ---------------- CPU samples ----------------
    df_cpu = new_trace_df(0)
    scs = None
    try:
        scs = cpu_mod.load_scs(logdir)
    except (ValueError, OSError) as e:
        p.print_warning(f"cpusamples.scs unreadable: {e}")
    if scs is not None:
        df_cpu = cpu_mod.scs_to_cputrace(scs, tb, logdir=logdir, symbolize=True)
        if scs.lost:
            p.print_warning(f"cpu sampler lost {scs.lost} records")
    if len(df_cpu):
        write_trace_csv(df_cpu, os.path.join(logdir, "cputrace.csv"))
        traces.append(
            SOFATrace(name="cpu_traces", title="CPU samples", color="DarkGray", data=df_cpu)
        )
        traces += _filter_traces(df_cpu, cfg.cpu_filters, "cpu")
        # optional swarm clustering (reference --enable_swarms,
        # bin/sofa_preprocess.py:1828-1836)
        if cfg.enable_swarms:
            try:
                from ..ml.hsg import hsg_cluster, swarms_to_traces

                swarms, captions = hsg_cluster(df_cpu, cfg.num_swarms, logdir)
                traces += swarms_to_traces(swarms, captions, logdir)
            except Exception as e:
                p.print_warning(f"swarm clustering failed: {e}")
    else:
        p.print_warning("no CPU samples")
    result["df_cpu"] = df_cpu
    if scs is not None and len(scs.samples_cs):
        try:
            from .flame import write_folded

            out = write_folded(scs, logdir)
            if out:
                p.print_info(f"flamegraph input written: {out}")
        except Exception as e:
            p.print_warning(f"stack folding failed: {e}")

    # ---------------- system monitors ----------------
    t_mp, mp_csv, usr_sys = sysmon.parse_mpstat(logdir, tb)
    if len(t_mp):
        traces.append(SOFATrace(name="mpstat_traces", title="CPU core busy (%)", color="CadetBlue", data=t_mp))
        mp_csv.to_csv(os.path.join(logdir, "mpstat.csv"), index=False)
        usr_sys.to_csv(os.path.join(logdir, "usr_sys.csv"), index=False)
    result["df_mpstat"] = t_mp

    t_disk, disk_vec = sysmon.parse_diskstat(logdir, tb)
    if len(t_disk):
        traces.append(SOFATrace(name="diskstat_traces", title="Disk throughput (MB/s)", color="SandyBrown", data=t_disk))
        disk_vec.to_csv(os.path.join(logdir, "diskstat_vector.csv"), index=False)
        # UI view: the most active device, or --diskstat_filters selection
        sel_dev = None
        if cfg.diskstat_filters:
            kw = cfg.diskstat_filters[0].keyword
            if (disk_vec["dev"] == kw).any():
                sel_dev = kw
        if sel_dev is None:
            totals = disk_vec.groupby("dev")[["read_Bps", "write_Bps"]].sum().sum(axis=1)
            sel_dev = totals.idxmax()
        disk_vec[disk_vec["dev"] == sel_dev].to_csv(
            os.path.join(logdir, "diskstat_vector_ui.csv"), index=False
        )
    result["df_diskstat"] = t_disk

    # per-request block-IO latency (tracefs stream; blktrace parity)
    try:
        from .blkio import parse_blkio

        df_blk = parse_blkio(logdir, tb)
        if len(df_blk):
            write_trace_csv(df_blk, os.path.join(logdir, "blktrace.csv"))
            traces.append(
                SOFATrace(name="blkio_traces", title="Block IO latency", color="Peru", data=df_blk)
            )
        result["df_blkio"] = df_blk
    except Exception as e:
        p.print_warning(f"blkio parse failed: {e}")
        result["df_blkio"] = new_trace_df(0)

    t_net, net_bw = sysmon.parse_netstat(logdir, tb)
    if len(t_net):
        traces.append(SOFATrace(name="netstat_traces", title="NIC throughput (MB/s)", color="YellowGreen", data=t_net))
        net_bw.to_csv(os.path.join(logdir, "netbandwidth.csv"), index=False)
        net_bw.to_csv(os.path.join(logdir, "netstat.csv"), index=False)
    result["df_netstat"] = t_net

    t_vm, vm_csv = sysmon.parse_vmstat(logdir, tb)
    if len(t_vm):
        traces.append(SOFATrace(name="vmstat_traces", title="Context switches/s", color="LightSteelBlue", data=t_vm))
        vm_csv.to_csv(os.path.join(logdir, "vmstat.csv"), index=False)
    result["df_vmstat"] = t_vm

    t_sm, t_gmem, gpusmi_csv = sysmon.parse_gpusmi(logdir, tb)
    if len(t_sm):
        traces.append(SOFATrace(name="gpusmi_sm_traces", title="GPU busy (%)", color="DarkOrange", data=t_sm))
        traces.append(SOFATrace(name="gpusmi_mem_traces", title="GPU mem/MM busy (%)", color="Gold", data=t_gmem))
        gpusmi_csv.to_csv(os.path.join(logdir, "gpusmi_trace.csv"), index=False)
    result["df_gpusmi"] = t_sm

    # measured per-xGMI-link bandwidth (gpu_metrics HW accumulators)
    try:
        t_xgmi, xgmi_csv = sysmon.parse_xgmi_counters(logdir, tb)
        if len(t_xgmi):
            traces.append(
                SOFATrace(name="xgmi_link_traces", title="xGMI link BW (GB/s)", color="MediumOrchid", data=t_xgmi)
            )
            xgmi_csv.to_csv(os.path.join(logdir, "xgmi_counters.csv"), index=False)
        result["df_xgmi"] = t_xgmi
    except Exception as e:
        p.print_warning(f"xgmi counter parse failed: {e}")
        result["df_xgmi"] = new_trace_df(0)

    # ---------------- GPU activity ----------------
    df_gpu = new_trace_df(0)
    df_rccl = new_trace_df(0)
    df_hip = new_trace_df(0)
    sgt_files = gpu_mod.load_sgt_files(logdir)
    if sgt_files:
        df_gpu = gpu_mod.sgt_to_gputrace(sgt_files, tb)
        df_rccl = gpu_mod.sgt_to_rccltrace(sgt_files, tb)
        df_hip = gpu_mod.sgt_to_hip_api_trace(sgt_files, tb)
        dropped = sum(s.dropped for s in sgt_files)
        if dropped:
            p.print_warning(f"GPU collector dropped {dropped} records")
    if len(df_gpu):
        write_trace_csv(df_gpu, os.path.join(logdir, "gputrace.csv"))
        traces.append(SOFATrace(name="gpu_traces", title="GPU kernels & copies", color="DarkSlateBlue", data=df_gpu))
        traces += _filter_traces(df_gpu, cfg.gpu_filters, "gpu")
    if len(df_rccl) == 0:
        # lite mode: collective args come from RCCL's own debug log
        try:
            from .rccl_log import parse_rccl_log

            df_rccl = parse_rccl_log(logdir)
        except Exception as e:
            p.print_warning(f"rccl debug-log parse failed: {e}")
    if len(df_rccl):
        write_trace_csv(df_rccl, os.path.join(logdir, "rccltrace.csv"))
        traces.append(SOFATrace(name="rccl_traces", title="RCCL collectives", color="Crimson", data=df_rccl))
    if len(df_hip):
        write_trace_csv(df_hip, os.path.join(logdir, "hip_api_trace.csv"))
        traces.append(SOFATrace(name="hip_api_traces", title="HIP API", color="MediumSeaGreen", data=df_hip))
    df_marks = gpu_mod.sgt_to_markers(sgt_files, tb) if sgt_files else new_trace_df(0)
    if len(df_marks):
        write_trace_csv(df_marks, os.path.join(logdir, "markers.csv"))
        traces.append(SOFATrace(name="roctx_traces", title="roctx markers", color="Black", data=df_marks))
    result["df_markers"] = df_marks
    # GPU PC samples (opt-in --pc_sampling; sdk collector)
    try:
        df_pc = gpu_mod.sgt_to_pcsamples(sgt_files, tb) if sgt_files else None
        if df_pc is not None and len(df_pc):
            df_pc.to_csv(os.path.join(logdir, "pcsamples.csv"), index=False)
            p.print_info(f"{len(df_pc)} GPU PC samples -> pcsamples.csv")
        result["df_pcsamples"] = df_pc
    except Exception as e:
        p.print_warning(f"pc sample parse failed: {e}")
        result["df_pcsamples"] = None

    df_kfd = gpu_mod.sgt_to_kfdtrace(sgt_files, tb) if sgt_files else new_trace_df(0)
    if len(df_kfd):
        write_trace_csv(df_kfd, os.path.join(logdir, "kfdtrace.csv"))
        traces.append(SOFATrace(name="kfd_traces", title="KFD page events", color="DarkRed", data=df_kfd))
    result["df_kfd"] = df_kfd
    result["df_gpu"] = df_gpu
    result["df_rccl"] = df_rccl
    result["df_hip"] = df_hip
    result["sgt_files"] = sgt_files

    # ---------------- packets (optional sniffer output) ----------------
    try:
        from . import net as net_mod

        df_pkt = net_mod.parse_pktcap(logdir, tb, cfg)
        if len(df_pkt):
            write_trace_csv(df_pkt, os.path.join(logdir, "nettrace.csv"))
            traces.append(SOFATrace(name="net_traces", title="Network packets", color="OliveDrab", data=df_pkt))
            traces += _filter_traces(df_pkt, cfg.net_filters, "net")
        result["df_net"] = df_pkt
    except Exception as e:
        p.print_warning(f"packet parse failed: {e}")
        result["df_net"] = new_trace_df(0)

    # ---------------- syscalls (SST from sofa-syscalltrace) ----------------
    try:
        from .strace import parse_sst

        df_strace = parse_sst(logdir, tb, cfg)
        if len(df_strace):
            write_trace_csv(df_strace, os.path.join(logdir, "strace.csv"))
            traces.append(SOFATrace(name="strace_traces", title="Syscalls", color="DarkKhaki", data=df_strace))
        result["df_strace"] = df_strace
    except Exception as e:
        p.print_warning(f"syscall trace parse failed: {e}")
        result["df_strace"] = new_trace_df(0)

    # ---------------- python stacks ----------------
    try:
        from .pystacks import parse_pystacks

        df_py = parse_pystacks(logdir, tb)
        if len(df_py):
            write_trace_csv(df_py, os.path.join(logdir, "pystacks.csv"))
            traces.append(SOFATrace(name="pystacks_traces", title="Python stacks", color="SteelBlue", data=df_py))
        result["df_pystacks"] = df_py
    except Exception as e:
        p.print_warning(f"pystacks parse failed: {e}")
        result["df_pystacks"] = new_trace_df(0)

    # ---------------- report.js + chrome trace ----------------
    traces_to_json(traces, os.path.join(logdir, "report.js"), plot_ratio=cfg.plot_ratio)
    try:
        from ..viz.chrome_trace import write_chrome_trace

        ct = write_chrome_trace(logdir, result)
        if ct:
            p.print_info(f"chrome://tracing export: {ct}")
    except Exception as e:
        p.print_warning(f"chrome trace export failed: {e}")
    n_events = sum(len(t.data) for t in traces if t.data is not None)
    dt = __import__("time").perf_counter() - t_start
    p.print_progress(
        "preprocess done: %d series, %d points in %.2f s (%.0f events/s)"
        % (len(traces), n_events, dt, n_events / max(dt, 1e-9))
    )
    result["traces"] = traces
    return result
