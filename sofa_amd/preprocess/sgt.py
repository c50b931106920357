"""Parser for the GPU collector binary format (SGT1).

Mirrors sofa_amd/native/collector/sgt_format.h (writer: sofatracer.cc).
Fixed-size record kinds are parsed with numpy run-detection (maximal
homogeneous runs viewed as structured arrays) — the events/sec hot path that
replaces the reference's nvprof-CSV + per-row Python parsing
(bin/sofa_preprocess.py:1420-1432).
"""

from __future__ import annotations

import struct
from dataclasses import dataclass, field
from typing import Dict, List

import numpy as np

MAGIC = 0x31544753
HEADER_SIZE = 64
HEADER_FMT = "<IIIIQQQ"  # magic, version, pid, pad, realtime, mono_raw, rocp (then 24B reserved)

REC_KERNEL = 1
REC_COPY = 2
REC_HIPAPI = 3
REC_RCCL = 4
REC_KERNEL_NAME = 5
REC_OPNAME = 6
REC_AGENT = 7
REC_CLOCK = 8
REC_ALLOC = 9
REC_DROP = 10
REC_MARKER = 11
REC_KFD = 12
REC_PCSAMPLE = 13

_HDR = [("type", "<u2"), ("size", "<u2"), ("_pad", "<u4")]

KERNEL_DTYPE = np.dtype(
    _HDR
    + [
        ("start_ns", "<u8"),
        ("end_ns", "<u8"),
        ("corr_id", "<u8"),
        ("tid", "<u4"),
        ("device", "<u4"),
        ("queue_id", "<u8"),
        ("kernel_id", "<u8"),
        ("private_segment_size", "<u4"),
        ("group_segment_size", "<u4"),
        ("grid_x", "<u4"),
        ("grid_y", "<u4"),
        ("grid_z", "<u4"),
        ("wg_x", "<u4"),
        ("wg_y", "<u4"),
        ("wg_z", "<u4"),
        ("_pad2", "<u4"),
        ("_pad3", "<u4"),
    ]
)

COPY_DTYPE = np.dtype(
    _HDR
    + [
        ("start_ns", "<u8"),
        ("end_ns", "<u8"),
        ("corr_id", "<u8"),
        ("tid", "<u4"),
        ("op", "<u4"),
        ("src_device", "<i4"),
        ("dst_device", "<i4"),
        ("bytes", "<u8"),
    ]
)

API_DTYPE = np.dtype(
    _HDR
    + [
        ("start_ns", "<u8"),
        ("end_ns", "<u8"),
        ("corr_id", "<u8"),
        ("tid", "<u4"),
        ("op", "<u4"),
    ]
)

RCCL_DTYPE = np.dtype(
    _HDR
    + [
        ("start_ns", "<u8"),
        ("end_ns", "<u8"),
        ("corr_id", "<u8"),
        ("tid", "<u4"),
        ("op", "<u4"),
        ("count", "<u8"),
        ("datatype", "<u4"),
        ("elem_size", "<u4"),
        ("peer_or_root", "<i4"),
        ("device", "<u4"),
        ("comm", "<u8"),
        ("stream", "<u8"),
    ]
)

KFD_DTYPE = np.dtype(
    _HDR
    + [
        ("timestamp", "<u8"),
        ("op_class", "<u4"),
        ("operation", "<u4"),
        ("pid", "<u4"),
        ("device", "<i4"),
        ("addr_start", "<u8"),
        ("addr_end", "<u8"),
        ("src_device", "<i4"),
        ("error_code", "<i4"),
    ]
)

ALLOC_DTYPE = np.dtype(
    _HDR
    + [
        ("start_ns", "<u8"),
        ("end_ns", "<u8"),
        ("corr_id", "<u8"),
        ("tid", "<u4"),
        ("op", "<u4"),
        ("device", "<i4"),
        ("_pad2", "<u4"),
        ("address", "<u8"),
        ("bytes", "<u8"),
    ]
)

PCSAMPLE_DTYPE = np.dtype(
    _HDR
    + [
        ("timestamp", "<u8"),
        ("corr_id", "<u8"),
        ("code_object_id", "<u8"),
        ("offset", "<u8"),
        ("exec_mask", "<u8"),
        ("dispatch_id", "<u8"),
        ("wave_in_group", "<u4"),
        ("device", "<u4"),
    ]
)

FIXED_DTYPES = {
    REC_KERNEL: KERNEL_DTYPE,
    REC_PCSAMPLE: PCSAMPLE_DTYPE,
    REC_KFD: KFD_DTYPE,
    REC_COPY: COPY_DTYPE,
    REC_HIPAPI: API_DTYPE,
    REC_RCCL: RCCL_DTYPE,
    REC_ALLOC: ALLOC_DTYPE,
}


@dataclass
class SgtFile:
    pid: int = 0
    realtime_ns: int = 0
    monotonic_raw_ns: int = 0
    rocp_ns: int = 0
    kernels: np.ndarray = field(default_factory=lambda: np.empty(0, KERNEL_DTYPE))
    copies: np.ndarray = field(default_factory=lambda: np.empty(0, COPY_DTYPE))
    hip_api: np.ndarray = field(default_factory=lambda: np.empty(0, API_DTYPE))
    rccl: np.ndarray = field(default_factory=lambda: np.empty(0, RCCL_DTYPE))
    allocs: np.ndarray = field(default_factory=lambda: np.empty(0, ALLOC_DTYPE))
    kfd: np.ndarray = field(default_factory=lambda: np.empty(0, KFD_DTYPE))
    pcsamples: np.ndarray = field(default_factory=lambda: np.empty(0, PCSAMPLE_DTYPE))
    kernel_names: Dict[int, str] = field(default_factory=dict)
    markers: List[tuple] = field(default_factory=list)  # (rocp_ns, message)
    opnames: Dict[tuple, str] = field(default_factory=dict)  # (kind, op) -> name
    agents: List[dict] = field(default_factory=list)
    clocks: List[tuple] = field(default_factory=list)  # (realtime, mono_raw, rocp)
    dropped: int = 0

    @property
    def n_events(self) -> int:
        return (
            len(self.kernels)
            + len(self.copies)
            + len(self.hip_api)
            + len(self.rccl)
            + len(self.allocs)
        )

    def rocp_to_mono_raw_offset(self) -> int:
        """Offset such that mono_raw_ns = rocp_ns + offset (first clock pair)."""
        if self.clocks:
            rt, mono, rocp = self.clocks[0]
            return mono - rocp
        return self.monotonic_raw_ns - self.rocp_ns

    def rocp_to_realtime_offset(self) -> int:
        if self.clocks:
            rt, mono, rocp = self.clocks[0]
            return rt - rocp
        return self.realtime_ns - self.rocp_ns


def parse_sgt(path: str) -> SgtFile:
    with open(path, "rb") as f:
        buf = f.read()
    out = SgtFile()
    if len(buf) < HEADER_SIZE:
        return out
    magic, version, pid, _pad, rt, mono, rocp = struct.unpack_from(HEADER_FMT, buf, 0)
    if magic != MAGIC:
        raise ValueError(f"{path}: bad SGT magic {magic:#x}")
    out.pid = pid
    out.realtime_ns = rt
    out.monotonic_raw_ns = mono
    out.rocp_ns = rocp

    chunks: Dict[int, List[np.ndarray]] = {k: [] for k in FIXED_DTYPES}
    off = HEADER_SIZE
    n = len(buf)
    u16 = np.frombuffer(buf[: n & ~1], dtype="<u2")
    while off + 8 <= n:
        rtype, rsize = struct.unpack_from("<HH", buf, off)
        if rsize == 0:
            break
        if off + rsize > n:
            break  # truncated tail (e.g. killed process)
        dtype = FIXED_DTYPES.get(rtype)
        if dtype is not None and rsize == dtype.itemsize:
            item = dtype.itemsize
            max_k = (n - off) // item
            base = off // 2
            stride = item // 2
            types = u16[base : base + max_k * stride : stride]
            sizes = u16[base + 1 : base + 1 + max_k * stride : stride]
            bad = np.nonzero((types != rtype) | (sizes != item))[0]
            k = int(bad[0]) if len(bad) else max_k
            arr = np.frombuffer(buf, dtype=dtype, count=k, offset=off)
            chunks[rtype].append(arr)
            off += k * item
        elif rtype == REC_KERNEL_NAME or rtype == REC_MARKER:
            (ident,) = struct.unpack_from("<Q", buf, off + 8)
            name = buf[off + 16 : off + rsize].split(b"\0", 1)[0].decode("utf-8", "replace")
            if rtype == REC_KERNEL_NAME:
                out.kernel_names[ident] = name
            else:
                out.markers.append((ident, name))
            off += rsize
        elif rtype == REC_OPNAME:
            kind, op = struct.unpack_from("<II", buf, off + 8)
            name = buf[off + 16 : off + rsize].split(b"\0", 1)[0].decode("utf-8", "replace")
            out.opnames[(kind, op)] = name
            off += rsize
        elif rtype == REC_AGENT:
            (handle, device, atype, node, wave, cu, xcc) = struct.unpack_from(
                "<QiiIIII", buf, off + 8
            )
            name = buf[off + 40 : off + 104].split(b"\0", 1)[0].decode("utf-8", "replace")
            out.agents.append(
                {
                    "handle": handle,
                    "device": device,
                    "type": atype,
                    "node_id": node,
                    "wave_front_size": wave,
                    "cu_count": cu,
                    "num_xcc": xcc,
                    "name": name,
                }
            )
            off += rsize
        elif rtype == REC_CLOCK:
            rt2, mono2, rocp2 = struct.unpack_from("<QQQ", buf, off + 8)
            out.clocks.append((rt2, mono2, rocp2))
            off += rsize
        elif rtype == REC_DROP:
            (dropped,) = struct.unpack_from("<Q", buf, off + 8)
            out.dropped += dropped
            off += rsize
        else:
            off += rsize

    def cat(key, dtype):
        c = chunks[key]
        if not c:
            return np.empty(0, dtype)
        return np.concatenate(c) if len(c) > 1 else c[0]

    out.kernels = cat(REC_KERNEL, KERNEL_DTYPE)
    out.copies = cat(REC_COPY, COPY_DTYPE)
    out.hip_api = cat(REC_HIPAPI, API_DTYPE)
    out.rccl = cat(REC_RCCL, RCCL_DTYPE)
    out.allocs = cat(REC_ALLOC, ALLOC_DTYPE)
    out.kfd = cat(REC_KFD, KFD_DTYPE)
    out.pcsamples = cat(REC_PCSAMPLE, PCSAMPLE_DTYPE)
    return out
