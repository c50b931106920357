"""sofa analyze — console report + features vector + hints.

Orchestration parity with reference bin/sofa_analyze.py:793-1055 and
cluster_analyze :1057-1137 (signature bug there NOT replicated).
Prints the `Complete!!` sentinel the e2e harness greps for
(reference test/test.py:72-74).
"""

from __future__ import annotations

import json
import os
import shutil
from typing import Dict, List, Optional, Tuple

import pandas as pd

from .. import printing as p
from ..config import SofaConfig
from ..schema import new_trace_df
from . import comm as comm_mod
from . import profiles
from .concurrency import concurrency_breakdown


def _load_csv_trace(logdir: str, name: str) -> pd.DataFrame:
    path = os.path.join(logdir, name)
    if not os.path.isfile(path):
        return new_trace_df(0)
    try:
        df = pd.read_csv(path)
    except Exception:
        p.print_warning(f"{name} unreadable; ignoring")
        return new_trace_df(0)
    # a corrupt/foreign CSV must not crash the profiles downstream
    required = {"timestamp", "duration", "name", "copyKind", "deviceId"}
    if not required <= set(df.columns):
        p.print_warning(f"{name} lacks trace columns; ignoring")
        return new_trace_df(0)
    return df


def _roi_filter(df: pd.DataFrame, begin: float, end: float) -> pd.DataFrame:
    if df is None or len(df) == 0 or end <= begin:
        return df
    return df[(df["timestamp"] >= begin) & (df["timestamp"] <= end)]


def sofa_analyze(cfg: SofaConfig, pre: Optional[dict] = None) -> Dict[str, float]:
    logdir = cfg.logdir
    p.print_title("SOFA analyze")
    features: List[Tuple[str, float]] = []

    pre = pre or {}
    df_cpu = pre.get("df_cpu")
    if df_cpu is None or len(df_cpu) == 0:
        df_cpu = _load_csv_trace(logdir, "cputrace.csv")
    df_gpu = pre.get("df_gpu")
    if df_gpu is None or len(df_gpu) == 0:
        df_gpu = _load_csv_trace(logdir, "gputrace.csv")
    df_rccl = pre.get("df_rccl")
    if df_rccl is None or len(df_rccl) == 0:
        df_rccl = _load_csv_trace(logdir, "rccltrace.csv")
    df_mpstat = pre.get("df_mpstat", new_trace_df(0))
    df_gpusmi = pre.get("df_gpusmi", new_trace_df(0))
    df_netstat = pre.get("df_netstat", new_trace_df(0))
    df_net = pre.get("df_net", new_trace_df(0))

    elapsed = 0.0
    misc_path = os.path.join(logdir, "misc.txt")
    if os.path.isfile(misc_path):
        try:
            with open(misc_path) as f:
                misc = json.load(f)
            elapsed = float(misc.get("elapsed_time", 0.0))
            features.append(("elapsed_time", elapsed))
        except (OSError, ValueError):
            pass

    # --- ROI (reference :873-896 spotlight + --profile_region) ---
    if cfg.profile_region:
        try:
            b, e = cfg.profile_region.split(",")
            cfg.roi_begin, cfg.roi_end = float(b), float(e)
        except ValueError:
            p.print_warning("bad --profile_region, expected begin,end seconds")
    elif cfg.spotlight_gpu and len(df_gpusmi):
        cfg.roi_begin, cfg.roi_end = profiles.spotlight_roi(df_gpusmi)
        if cfg.roi_end > cfg.roi_begin:
            p.print_hint(
                "spotlight ROI: %.2f..%.2f s (GPU-active region)" % (cfg.roi_begin, cfg.roi_end)
            )
    if cfg.roi_end > cfg.roi_begin:
        df_cpu = _roi_filter(df_cpu, cfg.roi_begin, cfg.roi_end)
        df_gpu = _roi_filter(df_gpu, cfg.roi_begin, cfg.roi_end)
        df_rccl = _roi_filter(df_rccl, cfg.roi_begin, cfg.roi_end)

    # --- xGMI ring recommendation (reference :825-869 + tools/xring.py) ---
    topo = comm_mod.load_topology(logdir)
    if topo and topo.get("n_gpus", 0) >= 2:
        rings = comm_mod.xgmi_rings(topo)
        if rings:
            ring = rings[0]
            hints_dir = os.path.join(logdir, "sofa_hints")
            os.makedirs(hints_dir, exist_ok=True)
            with open(os.path.join(hints_dir, "xring_order.txt"), "w") as f:
                f.write("export HIP_VISIBLE_DEVICES=%s\n" % ",".join(str(i) for i in ring))
            p.print_hint(
                "xGMI ring order: HIP_VISIBLE_DEVICES=%s (written to sofa_hints/xring_order.txt)"
                % ",".join(str(i) for i in ring)
            )

    # --- per-subsystem profiles ---
    profiles.cpu_profile(df_cpu, features, elapsed)
    profiles.mpstat_profile(df_mpstat, features, cfg.is_idle_threshold)
    profiles.vmstat_profile(logdir, features)
    profiles.diskstat_profile(logdir, features)
    df_blkio = pre.get("df_blkio")
    if df_blkio is None or len(df_blkio) == 0:
        df_blkio = _load_csv_trace(logdir, "blktrace.csv")
    profiles.blkio_latency_profile(df_blkio, features)
    profiles.netbandwidth_profile(logdir, features)
    profiles.net_profile(logdir, df_net, features)
    profiles.gpu_profile(df_gpu, df_rccl, features)
    profiles.gpusmi_profile(df_gpusmi, features, logdir)
    comm_mod.comm_profile(logdir, df_gpu, features)
    comm_mod.rccl_link_attribution(logdir, df_rccl, topo, features, df_gpu=df_gpu)
    profiles.xgmi_measured_profile(logdir, features)
    profiles.pc_hotspot_profile(logdir, features)

    # --- launch-latency / launch-bound analysis (corr-id join) ---
    try:
        from .launch import launch_latency_profile

        launch_latency_profile(pre.get("sgt_files") or [], features)
    except Exception as e:
        p.print_warning(f"launch-latency analysis failed: {e}")

    # --- clock-sync validation (timebase microkernel vs rocprofiler) ---
    tb_path = os.path.join(logdir, "gpu_timebase.json")
    sgt_files = pre.get("sgt_files") or []
    if os.path.isfile(tb_path) and sgt_files:
        try:
            with open(tb_path) as f:
                gtb = json.load(f)
            for sgt in sgt_files:
                if len(sgt.clocks) >= 2:
                    rt0, mono0, rocp0 = sgt.clocks[0]
                    rt1, mono1, rocp1 = sgt.clocks[-1]
                    span = mono1 - mono0
                    if span > 0:
                        drift_ppm = abs((rocp1 - rocp0) - (mono1 - mono0)) / span * 1e6
                        features.append(("clock_drift_ppm", float(drift_ppm)))
                        msg = (
                            "rocprofiler clock vs CLOCK_MONOTONIC_RAW drift: "
                            "%.1f ppm over %.2f s; device tick rate %.4f MHz "
                            "(s_memrealtime microkernel; raw clock carries no NTP rate corr.)"
                            % (drift_ppm, span * 1e-9, gtb.get("ticks_per_second", 0) / 1e6)
                        )
                        if drift_ppm > 500:
                            p.print_warning(msg)
                        else:
                            p.print_info(msg)
                    break
        except (OSError, ValueError, KeyError):
            pass
        # MFMA-timed marker: the tracer's span for mfma_marker_kernel vs the
        # kernel's own s_memrealtime measurement (on-device ground truth;
        # north-star "MFMA-timed markers" validation)
        try:
            with open(tb_path) as f:
                gtb = json.load(f)
            self_ns = gtb.get("mfma_marker_self_ns", 0.0)
            if self_ns > 0 and len(df_gpu):
                marks = df_gpu[
                    df_gpu["name"].astype(str).str.contains("mfma_marker_kernel")
                ]
                if len(marks):
                    traced_ns = float(marks["duration"].max()) * 1e9
                    err_pct = 100.0 * abs(traced_ns - self_ns) / self_ns
                    features.append(("mfma_marker_clock_err_pct", float(err_pct)))
                    msg = (
                        "MFMA-timed marker: tracer span %.1f us vs on-device "
                        "self-measurement %.1f us (%.2f%% error)"
                        % (traced_ns / 1e3, self_ns / 1e3, err_pct)
                    )
                    if err_pct > 5.0:
                        p.print_warning(msg)
                    else:
                        p.print_info(msg)
        except (OSError, ValueError, KeyError):
            pass

    # --- concurrency breakdown ---
    concurrency_breakdown(
        logdir, df_mpstat, df_gpusmi, df_netstat, features, window_s=1.0 / cfg.sys_mon_rate
    )

    # --- AISI (iteration detection) ---
    if cfg.enable_aisi:
        try:
            from ..aisi import sofa_aisi

            sofa_aisi(
                logdir, cfg, df_cpu, df_gpu, df_rccl, features,
                df_strace=pre.get("df_strace"),
            )
        except Exception as e:
            p.print_warning(f"AISI failed: {e}")

    # --- features dump (reference :993-999) ---
    fdf = pd.DataFrame(features, columns=["name", "value"])
    fdf.to_csv(os.path.join(logdir, "features.csv"), index=False)
    print("\nPerformance features:")
    print(fdf.to_string(index=False))

    # --- advisor (POTATO-parity gRPC + built-in rule engine,
    #     reference :1007-1048) ---
    try:
        from ..advisor.client import get_hint, local_hints

        hint = None
        if cfg.potato_server:
            try:
                hint = get_hint(cfg.potato_server, fdf)
            except Exception as e:
                p.print_warning(f"advisor server unavailable ({e}); using local rules")
        if hint is None:
            hint = local_hints(fdf)
        if hint:
            print("\nAdvisor hints:")
            p.print_hint(hint)
            with open(os.path.join(logdir, "potato_report.html"), "w") as f:
                f.write("<html><body><pre>%s</pre></body></html>" % hint)
    except Exception as e:
        p.print_warning(f"advisor failed: {e}")

    # --- copy sofaboard into logdir (reference :1050-1052) ---
    board_src = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "sofaboard")
    if os.path.isdir(board_src):
        for name in os.listdir(board_src):
            try:
                shutil.copyfile(os.path.join(board_src, name), os.path.join(logdir, name))
            except OSError:
                pass

    print("Complete!!")
    return dict(features)


def cluster_analyze(cfg: SofaConfig, node_cfgs: Dict[str, SofaConfig]) -> None:
    """Merge per-node logdirs <logdir>-<ip>/ (reference :1057-1137)."""
    p.print_title("SOFA cluster analyze")
    rows = []
    for ip, c2 in node_cfgs.items():
        feat_path = os.path.join(c2.logdir, "features.csv")
        feats = {}
        if os.path.isfile(feat_path):
            try:
                fdf = pd.read_csv(feat_path)
                feats = dict(zip(fdf["name"], fdf["value"]))
            except (OSError, ValueError):
                pass
        else:
            # analyze the node now to produce features
            feats = sofa_analyze(c2, {})
        rows.append(
            {
                "node": ip,
                "elapsed": feats.get("elapsed_time", 0.0),
                "cpu_busy": feats.get("cpu_mean_busy", 0.0),
                "gpu_time": feats.get("gpu_time", 0.0),
                "rccl_time": feats.get("rccl_time", feats.get("rccl_kernel_time", 0.0)),
                "net_tx_q50": feats.get("net_tx_q50", 0.0),
                "net_rx_q50": feats.get("net_rx_q50", 0.0),
            }
        )
    cdf = pd.DataFrame(rows)
    base = cfg.logdir.rstrip("/")
    os.makedirs(base, exist_ok=True)
    cdf.to_csv(os.path.join(base, "cluster_report.csv"), index=False)
    print(cdf.to_string(index=False))
    print("Complete!!")
