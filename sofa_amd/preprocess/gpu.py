"""GPU trace -> unified schema (gputrace.csv, rccltrace.csv, hip_api_trace.csv).

Replaces the reference's nvprof/nvvp pipeline (bin/sofa_preprocess.py:
1340-1543): SGT binary records from the collector are converted to the
13-column schema fully vectorized.

copyKind mapping keeps the reference's codes (bin/sofa_preprocess.py:294-326,
bin/sofa_common.py:20-21): 0 kernel, 1 H2D, 2 D2H, 8 D2D, 10 P2P, 16 RCCL.
"""

from __future__ import annotations

import glob
import os
from typing import List, Optional, Tuple

import numpy as np
import pandas as pd

from ..config import SofaConfig
from ..schema import new_trace_df
from .sgt import SgtFile, parse_sgt
from .symbols import demangle
from .timebase import TimeBase

# rocprofiler_memory_copy_operation_t -> (copyKind, label)
COPY_OP_MAP = {
    0: (8, "CopyUnknown"),
    1: (0, "CopyHostToHost"),
    2: (1, "CopyHostToDevice"),
    3: (2, "CopyDeviceToHost"),
    4: (8, "CopyDeviceToDevice"),
}

RCCL_COLL_NAMES = None  # resolved from opnames table per file


def load_sgt_files(logdir: str) -> List[SgtFile]:
    out = []
    for path in sorted(glob.glob(os.path.join(logdir, "gputrace_*.sgt"))):
        try:
            out.append(parse_sgt(path))
        except (ValueError, OSError) as e:
            from .. import printing as p

            p.print_warning(f"skipping {path}: {e}")
    return out


def _timeline(tb: Optional[TimeBase], sgt: SgtFile, ns: np.ndarray) -> np.ndarray:
    """rocprofiler ns -> timeline seconds (via the per-file clock pair)."""
    off = sgt.rocp_to_realtime_offset()
    epoch_s = (ns.astype(np.float64) + off) * 1e-9
    if tb is None:
        return epoch_s
    return epoch_s - tb.time_base


def sgt_to_gputrace(
    files: List[SgtFile], tb: Optional[TimeBase], demangle_names: bool = True
) -> pd.DataFrame:
    """Kernel + copy records -> unified trace rows."""
    frames = []
    for sgt in files:
        k = sgt.kernels
        if len(k):
            df = new_trace_df(len(k))
            ts = _timeline(tb, sgt, k["start_ns"])
            dur = (k["end_ns"] - k["start_ns"]).astype(np.float64) * 1e-9
            df["timestamp"] = ts
            df["duration"] = dur
            df["deviceId"] = k["device"].astype(np.int64)
            df["copyKind"] = 0
            df["pid"] = sgt.pid
            df["tid"] = k["tid"].astype(np.int64)
            df["event"] = k["kernel_id"].astype(np.float64)
            names = sgt.kernel_names
            if demangle_names:
                resolved = {kid: demangle(nm) for kid, nm in names.items()}
            else:
                resolved = names
            dev = k["device"]
            kid = k["kernel_id"]
            df["name"] = [
                "[gpu%d] %s" % (d, resolved.get(i, "kernel_%d" % i))
                for d, i in zip(dev, kid)
            ]
            df["category"] = 0
            frames.append(df)
        c = sgt.copies
        if len(c):
            df = new_trace_df(len(c))
            ts = _timeline(tb, sgt, c["start_ns"])
            dur = (c["end_ns"] - c["start_ns"]).astype(np.float64) * 1e-9
            op = c["op"].astype(np.int64)
            src = c["src_device"].astype(np.int64)
            dst = c["dst_device"].astype(np.int64)
            bytes_ = c["bytes"].astype(np.int64)
            ck = np.zeros(len(c), dtype=np.int64)
            labels = []
            for i in range(len(c)):
                k2, lbl = COPY_OP_MAP.get(int(op[i]), (8, "CopyUnknown"))
                # D2D across devices over xGMI = P2P (copyKind 10)
                if k2 == 8 and src[i] >= 0 and dst[i] >= 0 and src[i] != dst[i]:
                    k2 = 10
                    lbl = "CopyPeerToPeer"
                ck[i] = k2
                labels.append(
                    "[gpu%d] %s %d bytes (gpu%d->gpu%d)"
                    % (max(dst[i], src[i], 0), lbl, bytes_[i], src[i], dst[i])
                )
            df["timestamp"] = ts
            df["duration"] = dur
            df["deviceId"] = np.maximum(np.maximum(src, dst), 0)
            df["copyKind"] = ck
            df["payload"] = bytes_
            with np.errstate(divide="ignore", invalid="ignore"):
                bw = np.where(dur > 0, bytes_ / np.maximum(dur, 1e-12), 0.0)
            df["bandwidth"] = bw
            df["pkt_src"] = np.maximum(src, -1)
            df["pkt_dst"] = np.maximum(dst, -1)
            df["pid"] = sgt.pid
            df["tid"] = c["tid"].astype(np.int64)
            df["name"] = labels
            df["category"] = 0
            frames.append(df)
    if not frames:
        return new_trace_df(0)
    out = pd.concat(frames, ignore_index=True)
    out.sort_values("timestamp", inplace=True, kind="stable")
    out.reset_index(drop=True, inplace=True)
    return out


def sgt_to_rccltrace(files: List[SgtFile], tb: Optional[TimeBase]) -> pd.DataFrame:
    """RCCL API calls -> unified trace rows (copyKind 16).

    payload = count * elem_size (bytes at the API level); per-xGMI-link
    attribution happens in analyze.comm using topology + algorithm model.
    """
    frames = []
    for sgt in files:
        r = sgt.rccl
        if not len(r):
            continue
        df = new_trace_df(len(r))
        ts = _timeline(tb, sgt, r["start_ns"])
        dur = (r["end_ns"] - r["start_ns"]).astype(np.float64) * 1e-9
        payload = (r["count"] * r["elem_size"]).astype(np.int64)
        # opnames stores (kind,op); the RCCL kind enum value varies across SDK
        # versions, so filter by name prefix instead of kind.
        rccl_names = {}
        for (kind, op), nm in sgt.opnames.items():
            if nm.startswith("nccl") or nm.startswith("mscclpp"):
                rccl_names[op] = nm
        op = r["op"]
        df["timestamp"] = ts
        df["duration"] = dur
        df["deviceId"] = r["device"].astype(np.int64)
        df["copyKind"] = 16
        df["payload"] = payload
        with np.errstate(divide="ignore", invalid="ignore"):
            df["bandwidth"] = np.where(dur > 0, payload / np.maximum(dur, 1e-12), 0.0)
        df["pid"] = sgt.pid
        df["tid"] = r["tid"].astype(np.int64)
        df["event"] = op.astype(np.float64)
        df["pkt_src"] = r["comm"].astype(np.int64) & 0x7FFFFFFF  # comm identity
        df["pkt_dst"] = r["peer_or_root"].astype(np.int64)
        df["name"] = [
            "%s(count=%d, dtype=%d, comm=%x, stream=%x)"
            % (rccl_names.get(int(o), "rccl_op_%d" % o), cnt, dt, cm, st)
            for o, cnt, dt, cm, st in zip(
                op, r["count"], r["datatype"], r["comm"], r["stream"]
            )
        ]
        df["category"] = 0
        frames.append(df)
    if not frames:
        return new_trace_df(0)
    out = pd.concat(frames, ignore_index=True)
    out.sort_values("timestamp", inplace=True, kind="stable")
    out.reset_index(drop=True, inplace=True)
    return out


def sgt_to_hip_api_trace(files: List[SgtFile], tb: Optional[TimeBase]) -> pd.DataFrame:
    frames = []
    for sgt in files:
        a = sgt.hip_api
        if not len(a):
            continue
        df = new_trace_df(len(a))
        ts = _timeline(tb, sgt, a["start_ns"])
        dur = (a["end_ns"] - a["start_ns"]).astype(np.float64) * 1e-9
        hip_names = {
            op: nm
            for (kind, op), nm in sgt.opnames.items()
            if nm.startswith("hip")
        }
        op = a["op"]
        df["timestamp"] = ts
        df["duration"] = dur
        df["event"] = op.astype(np.float64)
        df["pid"] = sgt.pid
        df["tid"] = a["tid"].astype(np.int64)
        df["name"] = [hip_names.get(int(o), "hip_api_%d" % o) for o in op]
        df["category"] = 1
        frames.append(df)
    if not frames:
        return new_trace_df(0)
    out = pd.concat(frames, ignore_index=True)
    out.sort_values("timestamp", inplace=True, kind="stable")
    out.reset_index(drop=True, inplace=True)
    return out
