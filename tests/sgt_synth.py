"""Synthetic SGT writer — Python mirror of the collector's binary format
(sofa_amd/native/collector/sgt_format.h) used to test preprocess/analyze
GPU-free (the "fake trace-source layer" SURVEY.md §4 calls for)."""

import struct

import numpy as np

from sofa_amd.preprocess.sgt import (
    ALLOC_DTYPE,
    API_DTYPE,
    COPY_DTYPE,
    KERNEL_DTYPE,
    RCCL_DTYPE,
    REC_AGENT,
    REC_CLOCK,
    REC_COPY,
    REC_HIPAPI,
    REC_KERNEL,
    REC_KERNEL_NAME,
    REC_OPNAME,
    REC_PCSAMPLE,
    REC_RCCL,
)


class SgtWriter:
    def __init__(self, pid=1234, realtime_ns=10**18, monotonic_raw_ns=5 * 10**14, rocp_ns=10**12):
        self.buf = bytearray()
        hdr = struct.pack(
            "<IIIIQQQ", 0x31544753, 1, pid, 0, realtime_ns, monotonic_raw_ns, rocp_ns
        )
        self.buf += hdr + b"\0" * (64 - len(hdr))
        self.realtime_ns = realtime_ns
        self.rocp_ns = rocp_ns

    def clock(self, realtime_ns=None, mono=None, rocp=None):
        self.buf += struct.pack(
            "<HHIQQQ",
            REC_CLOCK,
            32,
            0,
            realtime_ns or self.realtime_ns,
            mono or 0,
            rocp or self.rocp_ns,
        )

    def agent(self, handle, device, atype=2, name="gfx950"):
        rec = struct.pack("<HHIQiiIIII", REC_AGENT, 104, 0, handle, device, atype, 0, 64, 256, 8)
        nm = name.encode()[:63]
        rec += nm + b"\0" * (64 - len(nm))
        self.buf += rec

    def kernel_name(self, kid, name):
        nm = name.encode() + b"\0"
        total = (16 + len(nm) + 7) & ~7
        self.buf += struct.pack("<HHIQ", REC_KERNEL_NAME, total, 0, kid) + nm + b"\0" * (
            total - 16 - len(nm)
        )

    def marker(self, rocp_ns, msg):
        nm = msg.encode() + b"\0"
        total = (16 + len(nm) + 7) & ~7
        self.buf += struct.pack("<HHIQ", 11, total, 0, rocp_ns) + nm + b"\0" * (
            total - 16 - len(nm)
        )

    def opname(self, kind, op, name):
        nm = name.encode() + b"\0"
        total = (16 + len(nm) + 7) & ~7
        self.buf += struct.pack("<HHIII", REC_OPNAME, total, 0, kind, op) + nm + b"\0" * (
            total - 16 - len(nm)
        )

    def kernel(self, start, end, kid, device=0, tid=1, corr=0, grid=(1024, 1, 1), wg=(256, 1, 1), lds=0):
        a = np.zeros(1, KERNEL_DTYPE)
        a["type"] = REC_KERNEL
        a["size"] = KERNEL_DTYPE.itemsize
        a["start_ns"] = start
        a["end_ns"] = end
        a["corr_id"] = corr
        a["tid"] = tid
        a["device"] = device
        a["kernel_id"] = kid
        a["group_segment_size"] = lds
        a["grid_x"], a["grid_y"], a["grid_z"] = grid
        a["wg_x"], a["wg_y"], a["wg_z"] = wg
        self.buf += a.tobytes()

    def copy(self, start, end, op, nbytes, src=-1, dst=0, tid=1):
        a = np.zeros(1, COPY_DTYPE)
        a["type"] = REC_COPY
        a["size"] = COPY_DTYPE.itemsize
        a["start_ns"] = start
        a["end_ns"] = end
        a["tid"] = tid
        a["op"] = op
        a["src_device"] = src
        a["dst_device"] = dst
        a["bytes"] = nbytes
        self.buf += a.tobytes()

    def hip_api(self, start, end, op, tid=1, corr=0):
        a = np.zeros(1, API_DTYPE)
        a["type"] = REC_HIPAPI
        a["size"] = API_DTYPE.itemsize
        a["start_ns"] = start
        a["end_ns"] = end
        a["corr_id"] = corr
        a["tid"] = tid
        a["op"] = op
        self.buf += a.tobytes()

    def rccl(self, start, end, op, count, elem_size=2, datatype=9, device=0, comm=0xABC, stream=0x1, peer=-1):
        a = np.zeros(1, RCCL_DTYPE)
        a["type"] = REC_RCCL
        a["size"] = RCCL_DTYPE.itemsize
        a["start_ns"] = start
        a["end_ns"] = end
        a["op"] = op
        a["count"] = count
        a["datatype"] = datatype
        a["elem_size"] = elem_size
        a["peer_or_root"] = peer
        a["device"] = device
        a["comm"] = comm
        a["stream"] = stream
        self.buf += a.tobytes()

    def pcsample(self, ts, corr, code_object_id=1, offset=0x100,
                 exec_mask=(1 << 64) - 1, dispatch_id=0, wave=0, device=0):
        self.buf += struct.pack(
            "<HHIQQQQQQII", REC_PCSAMPLE, 64, 0, ts, corr, code_object_id,
            offset, exec_mask, dispatch_id, wave, device,
        )

    def write(self, path):
        with open(path, "wb") as f:
            f.write(bytes(self.buf))
