"""Rule-based performance advisor.

The reference's POTATO path ships only the gRPC protocol (bin/potato_pb2*.py)
— the advisor server itself is external and absent from the repo (SURVEY.md
§2, L4d).  Here the rule engine is part of the framework, usable locally and
served over gRPC (advisor/server.py), tuned for MI355X systems.

Input: the features dict produced by sofa_analyze (name -> value).
Output: list of (metric, observation, suggestion) tuples.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

Hint = Tuple[str, str, str]  # metric, observation, suggestion


def advise(f: Dict[str, float]) -> List[Hint]:
    hints: List[Hint] = []
    g = f.get

    elapsed = g("elapsed_time", 0.0)

    # --- dominant-resource level ---
    if g("dominant_iow_ratio", 0.0) > 0.3:
        hints.append(
            (
                "dominant_iow_ratio",
                "run is IO-wait dominated (%.0f%% of active windows)" % (100 * g("dominant_iow_ratio", 0)),
                "stage the dataset on local NVMe or increase dataloader prefetch; "
                "check diskstat await in disk.html",
            )
        )
    if g("dominant_sys_ratio", 0.0) > 0.5:
        hints.append(
            (
                "dominant_sys_ratio",
                "kernel (sys) time dominates CPU activity",
                "look for page-cache churn, excessive syscalls, or host<->device "
                "pinned-memory copies taking the memcpy path",
            )
        )

    # --- GPU utilization ---
    max_util = max((v for k, v in f.items() if k.endswith("_util_q50")), default=None)
    if max_util is not None and max_util < 50 and g("gpu_time", 0.0) > 0:
        hints.append(
            (
                "gpu_util_q50",
                "median GPU utilization is only %.0f%%" % max_util,
                "the GPUs are starving: raise per-GPU batch size (288 GB HBM3E "
                "leaves room), overlap input pipeline, or fuse small kernels "
                "(hipGraphs for launch-bound inner loops)",
            )
        )

    # --- copy vs compute ---
    gpu_time = g("gpu_time", 0.0)
    memcpy_time = g("gpu_memcpy_time", 0.0)
    if gpu_time > 0 and memcpy_time / gpu_time > 0.2:
        hints.append(
            (
                "gpu_memcpy_time",
                "copies are %.0f%% of GPU activity" % (100 * memcpy_time / gpu_time),
                "keep tensors resident on-device; batch H2D transfers; use pinned "
                "host buffers (H2D bandwidth check: comm-report.html)",
            )
        )
    h2d_bw = g("h2d_bw", 0.0)
    if h2d_bw > 0 and h2d_bw < 10.0:
        hints.append(
            (
                "h2d_bw",
                "H2D bandwidth %.1f GB/s is far below PCIe/pinned capability" % h2d_bw,
                "use hipHostMalloc/pinned staging buffers for input batches",
            )
        )

    # --- collectives ---
    iter_step = g("iter_step_time", 0.0)
    coll = g("iter_coll_time", 0.0)
    if iter_step > 0 and (coll + g("iter_copy_time", 0.0)) / iter_step > 0.15:
        hints.append(
            (
                "iter_coll_time",
                "communication is %.0f%% of the training step" % (100 * (coll + g("iter_copy_time", 0.0)) / iter_step),
                "xGMI ring all-reduce is single-link-bound (7 links x ~153 GB/s): "
                "raise DDP bucket_cap_mb for fewer larger all-reduces, enable "
                "gradient-as-bucket-view, overlap backward with collectives; see "
                "xlink_traffic.csv for the hot link",
            )
        )
    if g("gpu_underfill_time_ratio", 0.0) > 0.25:
        hints.append(
            (
                "gpu_underfill_time_ratio",
                "%.0f%% of kernel time is in launches with <256 workgroups"
                % (g("gpu_underfill_time_ratio") * 100),
                "MI355X has 256 CUs over 8 XCDs: batch more work per launch "
                "(bigger batch/fused ops) or capture the launch-bound loop "
                "in a hipGraph; per-kernel tuning cannot fill the chip here",
            )
        )
    if g("rccl_hot_link_bytes", 0.0) > 0:
        hints.append(
            (
                "rccl_hot_link_bytes",
                "per-link RCCL traffic attributed (xlink_traffic.csv)",
                "if one link dominates, check the ring order hint in "
                "sofa_hints/xring_order.txt (HIP_VISIBLE_DEVICES)",
            )
        )

    # --- CPU side ---
    if g("cpu_active_ratio", 0.0) > 0.9 and (max_util or 0) < 60:
        hints.append(
            (
                "cpu_active_ratio",
                "all CPU cores busy while GPUs are under-utilized",
                "CPU-side preprocessing is the bottleneck: move augmentation "
                "on-device or add dataloader workers",
            )
        )
    if g("ctxt_per_s", 0.0) > 200000:
        hints.append(
            (
                "ctxt_per_s",
                "very high context-switch rate (%.0f/s)" % g("ctxt_per_s", 0.0),
                "over-subscribed threads; pin OMP_NUM_THREADS and dataloader workers",
            )
        )

    # --- network ---
    if g("net_tx_max", 0.0) > 1e9 or g("net_rx_max", 0.0) > 1e9:
        hints.append(
            (
                "net_tx_max",
                "NIC traffic approaches %.1f GB/s" % (max(g("net_tx_max", 0.0), g("net_rx_max", 0.0)) / 1e9),
                "multi-node scaling will bottleneck here before xGMI does; "
                "consider gradient compression or hierarchical all-reduce",
            )
        )

    if not hints:
        hints.append(
            (
                "overall",
                "no obvious bottleneck in the feature vector",
                "drill into the timeline (index.html) and per-iteration profile "
                "(--enable_aisi) for finer-grained analysis",
            )
        )
    return hints


def format_hints(hints: List[Hint]) -> str:
    out = []
    for metric, obs, sug in hints:
        out.append(f"[{metric}]\n  observation: {obs}\n  suggestion:  {sug}")
    return "\n".join(out)
