"""ctypes binding for librocm_smi64 — GPU telemetry + xGMI topology.

MI355X-native replacement for the reference's `nvidia-smi dmon/query/topo`
subprocess polling (cyliustack/sofa bin/sofa_record.py:300-312): direct
in-process library calls at the poll rate, no subprocess per sample.

Gracefully degrades to a stub when no GPU / no driver is present (CPU-only
containers), matching the reference's "no-GPU degradation is the tested path"
property.
"""

from __future__ import annotations

import ctypes
import ctypes.util
from typing import List, Optional

RSMI_STATUS_SUCCESS = 0

# rsmi_utilization_counter_type
RSMI_COARSE_GRAIN_GFX_ACTIVITY = 0
RSMI_COARSE_GRAIN_MEM_ACTIVITY = 1

# rsmi_memory_type_t
RSMI_MEM_TYPE_VRAM = 0


class RocmSmi:
    """Thin wrapper; every getter returns None on failure."""

    def __init__(self, lib_path: str = "/opt/rocm/lib/librocm_smi64.so"):
        self.lib = None
        self.n_devices = 0
        try:
            self.lib = ctypes.CDLL(lib_path)
        except OSError:
            return
        try:
            if self.lib.rsmi_init(ctypes.c_uint64(0)) != RSMI_STATUS_SUCCESS:
                self.lib = None
                return
            n = ctypes.c_uint32(0)
            if self.lib.rsmi_num_monitor_devices(ctypes.byref(n)) == RSMI_STATUS_SUCCESS:
                self.n_devices = int(n.value)
        except Exception:
            self.lib = None

    @property
    def available(self) -> bool:
        return self.lib is not None and self.n_devices > 0

    def shutdown(self) -> None:
        if self.lib is not None:
            try:
                self.lib.rsmi_shut_down()
            except Exception:
                pass

    def busy_percent(self, dev: int) -> Optional[int]:
        if not self.lib:
            return None
        v = ctypes.c_uint32(0)
        if self.lib.rsmi_dev_busy_percent_get(dev, ctypes.byref(v)) == RSMI_STATUS_SUCCESS:
            return int(v.value)
        return None

    def memory_busy_percent(self, dev: int) -> Optional[int]:
        if not self.lib:
            return None
        v = ctypes.c_uint32(0)
        if self.lib.rsmi_dev_memory_busy_percent_get(dev, ctypes.byref(v)) == RSMI_STATUS_SUCCESS:
            return int(v.value)
        return None

    def memory_usage(self, dev: int) -> Optional[int]:
        if not self.lib:
            return None
        v = ctypes.c_uint64(0)
        if (
            self.lib.rsmi_dev_memory_usage_get(dev, RSMI_MEM_TYPE_VRAM, ctypes.byref(v))
            == RSMI_STATUS_SUCCESS
        ):
            return int(v.value)
        return None

    def power_watts(self, dev: int) -> Optional[float]:
        if not self.lib:
            return None
        v = ctypes.c_uint64(0)
        try:
            if self.lib.rsmi_dev_power_ave_get(dev, 0, ctypes.byref(v)) == RSMI_STATUS_SUCCESS:
                return v.value / 1e6  # microwatts
        except Exception:
            pass
        return None

    def link_weight(self, src: int, dst: int) -> Optional[int]:
        if not self.lib:
            return None
        v = ctypes.c_uint64(0)
        if self.lib.rsmi_topo_get_link_weight(src, dst, ctypes.byref(v)) == RSMI_STATUS_SUCCESS:
            return int(v.value)
        return None

    def link_type(self, src: int, dst: int) -> Optional[tuple]:
        """Returns (hops, type) where type 2=XGMI, 1=PCIe (RSMI_IOLINK_TYPE)."""
        if not self.lib:
            return None
        hops = ctypes.c_uint64(0)
        ltype = ctypes.c_int(0)
        if (
            self.lib.rsmi_topo_get_link_type(src, dst, ctypes.byref(hops), ctypes.byref(ltype))
            == RSMI_STATUS_SUCCESS
        ):
            return (int(hops.value), int(ltype.value))
        return None

    def minmax_bandwidth(self, src: int, dst: int) -> Optional[tuple]:
        """(min,max) link bandwidth in MB/s between two devices over xGMI."""
        if not self.lib:
            return None
        mn = ctypes.c_uint64(0)
        mx = ctypes.c_uint64(0)
        try:
            if (
                self.lib.rsmi_minmax_bandwidth_get(src, dst, ctypes.byref(mn), ctypes.byref(mx))
                == RSMI_STATUS_SUCCESS
            ):
                return (int(mn.value), int(mx.value))
        except Exception:
            pass
        return None

    # rsmi_gpu_metrics_t field offsets, verified against the installed
    # header with a C offsetof probe (rocm_smi.h:1051-1244, ROCm 7.2):
    # the driver fills structure_size at +0; fields beyond it stay zero.
    _GM_SIZE = 4544
    _GM_GFX = 16       # u16 average_gfx_activity (%)
    _GM_UMC = 18       # u16 average_umc_activity (%)
    _GM_MM = 20        # u16 average_mm_activity (%) — VCN enc/dec, the
    #                    nvsmi-dmon enc/dec analog (ref preprocess:1097-1183)
    _GM_VCN = 122      # u16[4] vcn_activity
    _GM_XGMI_R = 184   # u64[8] xgmi_read_data_acc (KB, accumulated)
    _GM_XGMI_W = 248   # u64[8] xgmi_write_data_acc
    _GM_JPEG = 352     # u16[32] jpeg_activity

    def gpu_metrics(self, dev: int) -> Optional[dict]:
        """Raw gpu_metrics table: media-engine busy + per-xGMI-link traffic
        accumulators (the HW counters behind the analytic ring model)."""
        if not self.available:
            return None
        import struct as _struct

        buf = (ctypes.c_uint8 * self._GM_SIZE)()
        fn = getattr(self.lib, "rsmi_dev_gpu_metrics_info_get", None)
        if fn is None or fn(dev, ctypes.byref(buf)) != RSMI_STATUS_SUCCESS:
            return None
        raw = bytes(buf)
        gfx, umc, mm = _struct.unpack_from("<HHH", raw, self._GM_GFX)
        vcn = _struct.unpack_from("<4H", raw, self._GM_VCN)
        xr = _struct.unpack_from("<8Q", raw, self._GM_XGMI_R)
        xw = _struct.unpack_from("<8Q", raw, self._GM_XGMI_W)
        jpeg = _struct.unpack_from("<32H", raw, self._GM_JPEG)
        invalid = 0xFFFF
        return {
            "gfx_activity": gfx if gfx != invalid else -1,
            "umc_activity": umc if umc != invalid else -1,
            "mm_activity": mm if mm != invalid else -1,
            "vcn_activity": [v if v != invalid else -1 for v in vcn],
            "jpeg_activity": [v if v != invalid else -1 for v in jpeg],
            "xgmi_read_kb": list(xr),
            "xgmi_write_kb": list(xw),
        }

    def topology_matrix(self) -> List[List[dict]]:
        """NxN matrix of {hops,type,weight,bw_min,bw_max} between GPU pairs."""
        n = self.n_devices
        out: List[List[dict]] = []
        for i in range(n):
            row = []
            for j in range(n):
                if i == j:
                    row.append({"hops": 0, "type": 0, "weight": 0})
                    continue
                lt = self.link_type(i, j) or (-1, -1)
                w = self.link_weight(i, j)
                bw = self.minmax_bandwidth(i, j)
                row.append(
                    {
                        "hops": lt[0],
                        "type": lt[1],
                        "weight": w if w is not None else -1,
                        "bw_min": bw[0] if bw else -1,
                        "bw_max": bw[1] if bw else -1,
                    }
                )
            out.append(row)
        return out
