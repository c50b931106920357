"""Python API for the device trace ring (native/hip/trace_ring.hip).

Usage pattern for users instrumenting their own HIP kernels with
`ring_push` (see trace_ring.hip device API):

    from sofa_amd.record.ring_dump import DeviceTraceRing
    ring = DeviceTraceRing(device=0, capacity=1 << 20)
    # ... pass ring.handle into your extension; device code ring_push()es ...
    n = ring.dump_sgt(logdir, tag_names=["load", "compute", "store"])
    # -> logdir/gputrace_ring_<pid>.sgt, merged by `sofa preprocess` onto the
    #    unified timeline as devring:<tag> rows.
"""

from __future__ import annotations

import ctypes
import os
from typing import List, Optional


def _lib():
    path = os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "native", "lib", "libsofahip.so",
    )
    lib = ctypes.CDLL(path)
    lib.sofa_ring_create.argtypes = [ctypes.c_int, ctypes.c_uint32, ctypes.POINTER(ctypes.c_void_p)]
    lib.sofa_ring_destroy.argtypes = [ctypes.c_void_p]
    lib.sofa_ring_head.argtypes = [ctypes.c_void_p, ctypes.POINTER(ctypes.c_ulonglong)]
    lib.sofa_ring_test_produce.argtypes = [ctypes.c_void_p, ctypes.c_uint32, ctypes.c_uint32]
    lib.sofa_ring_dump_sgt.argtypes = [
        ctypes.c_void_p, ctypes.c_int, ctypes.c_char_p,
        ctypes.POINTER(ctypes.c_char_p), ctypes.c_int,
    ]
    return lib


class DeviceTraceRing:
    def __init__(self, device: int = 0, capacity: int = 1 << 20):
        self._lib = _lib()
        self._ring = ctypes.c_void_p()
        self.device = device
        rc = self._lib.sofa_ring_create(device, capacity, ctypes.byref(self._ring))
        if rc != 0:
            raise RuntimeError("sofa_ring_create failed (no GPU?)")

    @property
    def handle(self) -> int:
        """Opaque ring pointer for passing into user HIP extensions."""
        return self._ring.value

    @property
    def head(self) -> int:
        h = ctypes.c_ulonglong(0)
        self._lib.sofa_ring_head(self._ring, ctypes.byref(h))
        return int(h.value)

    def test_produce(self, n: int, n_tags: int = 3) -> None:
        rc = self._lib.sofa_ring_test_produce(self._ring, n, n_tags)
        if rc != 0:
            raise RuntimeError("ring produce failed")

    def dump_sgt(self, logdir: str, tag_names: Optional[List[str]] = None) -> int:
        names = tag_names or ["event"]
        arr = (ctypes.c_char_p * len(names))(*[n.encode() for n in names])
        n = self._lib.sofa_ring_dump_sgt(
            self._ring, self.device, str(logdir).encode(), arr, len(names)
        )
        if n < 0:
            raise RuntimeError("ring dump failed")
        return int(n)

    def close(self) -> None:
        if self._ring:
            self._lib.sofa_ring_destroy(self._ring)
            self._ring = ctypes.c_void_p()

    def __del__(self):  # pragma: no cover
        try:
            self.close()
        except Exception:
            pass
