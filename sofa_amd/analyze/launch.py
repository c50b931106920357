"""Launch-latency analysis: join kernel dispatches with their hipLaunchKernel
API spans by rocprofiler correlation id.

Per kernel: latency = gpu_start - api_end (time from enqueue returning to the
kernel starting on device).  Large medians with small kernel durations =
launch-bound inner loop -> hipGraphs hint (256 CUs need >>256 workgroups AND
enough enqueued work to stay fed).  No reference analog (nvprof's API trace
was unused by sofa's analyzer).
"""

from __future__ import annotations

from typing import List, Tuple

import numpy as np

from .. import printing as p


def launch_latency_profile(sgt_files, features: List[Tuple[str, float]]) -> None:
    if not sgt_files:
        return
    lat_all = []
    gaps_all = []
    for sgt in sgt_files:
        k = sgt.kernels
        a = sgt.hip_api
        if not len(k) or not len(a):
            continue
        # api correlation_id -> end timestamp (launch-class ops only; the
        # filtered default op set is already launch/copy/sync/alloc)
        order = np.argsort(a["corr_id"])
        a_sorted = a[order]
        idx = np.searchsorted(a_sorted["corr_id"], k["corr_id"])
        idx = np.clip(idx, 0, len(a_sorted) - 1)
        matched = a_sorted["corr_id"][idx] == k["corr_id"]
        if matched.any() and k["corr_id"].max() > 0:
            lat = (
                k["start_ns"][matched].astype(np.int64)
                - a_sorted["end_ns"][idx[matched]].astype(np.int64)
            )
            lat = lat[lat >= 0]  # negative = kernel started before API returned
            if len(lat):
                lat_all.append(lat)
        else:
            # lite-collector traces: kernels carry no correlation ids; the
            # host-side aqlSubmitBatch spans (op 60000, corr_id = packets in
            # batch) are order-matched — expand each batch span over its
            # packet count and zip with kernels in start order (exact for
            # the common single-queue stream)
            subs = a[a["op"] == 60000]
            if len(subs):
                order_s = np.argsort(subs["start_ns"])
                ends = np.repeat(
                    subs["end_ns"][order_s].astype(np.int64),
                    np.maximum(subs["corr_id"][order_s].astype(np.int64), 1),
                )
                ks = np.sort(k, order="start_ns")
                n = min(len(ends), len(ks))
                lat = ks["start_ns"][:n].astype(np.int64) - ends[:n]
                lat = lat[lat >= 0]
                if len(lat):
                    lat_all.append(lat)
        # device idle gaps between consecutive kernels on one queue
        ks = np.sort(k, order="start_ns")
        gaps = ks["start_ns"][1:].astype(np.int64) - ks["end_ns"][:-1].astype(np.int64)
        gaps_all.append(np.clip(gaps, 0, None))

    if not lat_all:
        return
    lat = np.concatenate(lat_all)
    gaps = np.concatenate(gaps_all) if gaps_all else np.empty(0, np.int64)
    p50, p95 = np.percentile(lat, [50, 95]) / 1e3
    features.append(("launch_latency_us_p50", float(p50)))
    features.append(("launch_latency_us_p95", float(p95)))
    print("\nEnqueue-to-start delay: p50 %.1f us, p95 %.1f us over %d matched "
          "launches (includes enqueue-ahead backlog: large values in an "
          "unsynchronized loop mean the CPU runs ahead, not launch overhead)"
          % (p50, p95, len(lat)))
    if len(gaps):
        idle_ratio = float(gaps.sum()) / max(
            float(gaps.sum() + np.concatenate([(s.kernels["end_ns"] - s.kernels["start_ns"]) for s in sgt_files if len(s.kernels)]).sum()), 1.0
        )
        features.append(("gpu_idle_gap_ratio", float(idle_ratio)))
        if idle_ratio > 0.3:
            p.print_hint(
                "GPU idle between kernels %.0f%% of busy+idle time — the inner loop "
                "looks LAUNCH-BOUND: capture it in a hipGraph, batch small ops, or "
                "raise per-launch work (256 CUs want >>256 workgroups)" % (100 * idle_ratio)
            )
