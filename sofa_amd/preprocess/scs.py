"""Parser for the cpusampler binary format (SCS1).

Mirrors sofa_amd/native/cpusampler/cpusampler.cc (writer).  The hot path —
PERF_RECORD_SAMPLE records — is parsed with numpy run-detection: maximal runs
of fixed-48-byte sample records are viewed as one structured array (zero
Python per-row work), replacing the reference's `perf script` text pipeline +
mp.Pool per-row parsing (bin/sofa_preprocess.py:1791-1799).
"""

from __future__ import annotations

import struct
from dataclasses import dataclass, field
from typing import Dict, List, Tuple

import numpy as np

HEADER_SIZE = 72
HEADER_FMT = "<IIQQQII"  # magic, version, realtime, mono_raw, boottime, freq, ncpus
MAGIC = 0x31534353

REC_SAMPLE = 1
REC_MMAP = 2
REC_COMM = 3
REC_EXIT = 4
REC_LOST = 5
REC_SAMPLE_CS = 6

MAX_FRAMES = 16

SAMPLE_SIZE = 48
# RecHeader(4) + pad(4 by alignment)?  cpusampler.cc SampleRec:
#   {u16,u16} header, then u64 time at offset 8 (struct padded), u64 ip,
#   u32 pid, u32 tid, u32 cpu, u32 flags, u64 period  -> 48 bytes
SAMPLE_DTYPE = np.dtype(
    [
        ("type", "<u2"),
        ("size", "<u2"),
        ("_pad", "<u4"),
        ("time_ns", "<u8"),
        ("ip", "<u8"),
        ("pid", "<u4"),
        ("tid", "<u4"),
        ("cpu", "<u4"),
        ("flags", "<u4"),
        ("period", "<u8"),
    ]
)
assert SAMPLE_DTYPE.itemsize == SAMPLE_SIZE

# -g mode: fixed-depth callchain samples (cpusampler.cc SampleCsRec, 184 B)
SAMPLE_CS_DTYPE = np.dtype(
    [
        ("type", "<u2"),
        ("size", "<u2"),
        ("_pad", "<u4"),
        ("time_ns", "<u8"),
        ("ip", "<u8"),
        ("pid", "<u4"),
        ("tid", "<u4"),
        ("cpu", "<u4"),
        ("flags", "<u4"),
        ("period", "<u8"),
        ("n_frames", "<u4"),
        ("_pad2", "<u4"),
        ("frames", "<u8", (MAX_FRAMES,)),
    ]
)
assert SAMPLE_CS_DTYPE.itemsize == 184

MMAP_FIXED = 40  # RecHeader(4)+pad(4)+time(8)+pid(4)+tid(4)+addr(8)+len(8)+pgoff(8) = 48?


@dataclass
class ScsFile:
    realtime_ns: int = 0
    monotonic_raw_ns: int = 0
    boottime_ns: int = 0
    sample_freq: int = 0
    n_cpus: int = 0
    event_type: int = 0  # header reserved[0]: 0 = cpu-clock (ns), 1 = cycles
    samples: np.ndarray = field(default_factory=lambda: np.empty(0, dtype=SAMPLE_DTYPE))
    samples_cs: np.ndarray = field(default_factory=lambda: np.empty(0, dtype=SAMPLE_CS_DTYPE))
    # pid -> list of (addr, len, pgoff, filename)
    mmaps: Dict[int, List[Tuple[int, int, int, str]]] = field(default_factory=dict)
    comms: Dict[int, str] = field(default_factory=dict)
    lost: int = 0


def parse_scs(path: str) -> ScsFile:
    with open(path, "rb") as f:
        buf = f.read()
    out = ScsFile()
    if len(buf) < HEADER_SIZE:
        return out
    magic, version, rt, mono, boot, freq, ncpus = struct.unpack_from(HEADER_FMT, buf, 0)
    if magic != MAGIC:
        raise ValueError(f"{path}: bad SCS magic {magic:#x}")
    out.realtime_ns = rt
    out.monotonic_raw_ns = mono
    out.boottime_ns = boot
    out.sample_freq = freq
    out.n_cpus = ncpus
    # reserved[0] (u8 little-endian right after the packed header fields)
    if len(buf) >= struct.calcsize(HEADER_FMT) + 8:
        out.event_type = struct.unpack_from("<Q", buf, struct.calcsize(HEADER_FMT))[0]

    sample_chunks: List[np.ndarray] = []
    cs_chunks: List[np.ndarray] = []
    off = HEADER_SIZE
    n = len(buf)
    u16 = np.frombuffer(buf[: n & ~1], dtype="<u2")
    while off + 4 <= n:
        rtype, rsize = struct.unpack_from("<HH", buf, off)
        if rsize == 0:
            break
        if rtype == REC_SAMPLE and rsize == SAMPLE_SIZE:
            # vectorized run detection: how many consecutive 48-byte sample
            # records start here?
            max_k = (n - off) // SAMPLE_SIZE
            if max_k > 0:
                # type/size u16 pairs at stride 24 (u16 units)
                base = off // 2
                types = u16[base : base + max_k * 24 : 24]
                sizes = u16[base + 1 : base + 1 + max_k * 24 : 24]
                bad = np.nonzero((types != REC_SAMPLE) | (sizes != SAMPLE_SIZE))[0]
                k = int(bad[0]) if len(bad) else max_k
            else:
                k = 0
            if k == 0:
                off += rsize
                continue
            arr = np.frombuffer(buf, dtype=SAMPLE_DTYPE, count=k, offset=off)
            sample_chunks.append(arr)
            off += k * SAMPLE_SIZE
        elif rtype == REC_SAMPLE_CS and rsize == SAMPLE_CS_DTYPE.itemsize:
            item = SAMPLE_CS_DTYPE.itemsize
            max_k = (n - off) // item
            base = off // 2
            stride = item // 2
            types = u16[base : base + max_k * stride : stride]
            sizes = u16[base + 1 : base + 1 + max_k * stride : stride]
            bad = np.nonzero((types != REC_SAMPLE_CS) | (sizes != item))[0]
            k = int(bad[0]) if len(bad) else max_k
            arr = np.frombuffer(buf, dtype=SAMPLE_CS_DTYPE, count=k, offset=off)
            cs_chunks.append(arr)
            off += k * item
        elif rtype == REC_MMAP:
            (time_ns, pid, tid, addr, ln, pgoff) = struct.unpack_from("<QIIQQQ", buf, off + 8)
            name = buf[off + 8 + 40 : off + rsize].split(b"\0", 1)[0].decode("utf-8", "replace")
            out.mmaps.setdefault(pid, []).append((addr, ln, pgoff, name))
            off += rsize
        elif rtype == REC_COMM:
            (time_ns, pid, tid) = struct.unpack_from("<QII", buf, off + 8)
            name = buf[off + 8 + 16 : off + rsize].split(b"\0", 1)[0].decode("utf-8", "replace")
            out.comms[tid] = name
            off += rsize
        elif rtype == REC_LOST:
            (_t, lost) = struct.unpack_from("<QQ", buf, off + 8)
            out.lost += lost
            off += rsize
        else:
            off += rsize
    if sample_chunks:
        out.samples = np.concatenate(sample_chunks) if len(sample_chunks) > 1 else sample_chunks[0]
    if cs_chunks:
        out.samples_cs = np.concatenate(cs_chunks) if len(cs_chunks) > 1 else cs_chunks[0]
        # expose CS samples through the flat view too (shared field prefix)
        flat = np.zeros(len(out.samples_cs), dtype=SAMPLE_DTYPE)
        for f in ("time_ns", "ip", "pid", "tid", "cpu", "flags", "period"):
            flat[f] = out.samples_cs[f]
        flat["type"] = REC_SAMPLE
        flat["size"] = SAMPLE_SIZE
        out.samples = (
            np.concatenate([out.samples, flat]) if len(out.samples) else flat
        )
    return out
