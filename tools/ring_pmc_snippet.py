"""Compaction-kernel workload for rocprofv3 stats/PMC collection."""
import ctypes, os, sys
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
lib = ctypes.CDLL(os.path.join(REPO, "sofa_amd", "native", "lib", "libsofahip.so"))
lib.sofa_ring_create.argtypes = [ctypes.c_int, ctypes.c_uint32, ctypes.POINTER(ctypes.c_void_p)]
lib.sofa_ring_test_produce.argtypes = [ctypes.c_void_p, ctypes.c_uint32, ctypes.c_uint32]
lib.sofa_ring_compact_bench.argtypes = [ctypes.c_void_p, ctypes.c_uint64, ctypes.c_int, ctypes.POINTER(ctypes.c_double)]
r = ctypes.c_void_p()
assert lib.sofa_ring_create(0, 1 << 24, ctypes.byref(r)) == 0
assert lib.sofa_ring_test_produce(r, 1 << 24, 7) == 0
ms = ctypes.c_double(0)
assert lib.sofa_ring_compact_bench(r, 20, 5, ctypes.byref(ms)) == 0
print("kernel ms", ms.value)
