"""GPU numerics tests for the CDNA4 trace-ring compaction kernel.

The compaction output is compared against a CPU (numpy) reference of the same
filter + clock transform (the repo-wide numerics-test pattern: HIP kernel vs
plain host reference).
"""

import ctypes
import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(REPO, "sofa_amd", "native", "lib", "libsofahip.so")

REC_DTYPE = np.dtype(
    [
        ("t_start", "<u8"),
        ("t_end", "<u8"),
        ("tag", "<u4"),
        ("src", "<u4"),
        ("arg", "<u8"),
    ]
)
assert REC_DTYPE.itemsize == 32


@pytest.fixture(scope="module")
def lib():
    if not os.path.exists(LIB):
        pytest.skip("libsofahip.so not built")
    lib = ctypes.CDLL(LIB)
    lib.sofa_ring_create.argtypes = [ctypes.c_int, ctypes.c_uint32, ctypes.POINTER(ctypes.c_void_p)]
    lib.sofa_ring_destroy.argtypes = [ctypes.c_void_p]
    lib.sofa_ring_test_produce.argtypes = [ctypes.c_void_p, ctypes.c_uint32, ctypes.c_uint32]
    lib.sofa_ring_head.argtypes = [ctypes.c_void_p, ctypes.POINTER(ctypes.c_ulonglong)]
    lib.sofa_ring_compact.argtypes = [
        ctypes.c_void_p,
        ctypes.c_uint64,
        ctypes.c_double,
        ctypes.c_longlong,
        ctypes.c_void_p,
        ctypes.c_uint32,
        ctypes.POINTER(ctypes.c_uint32),
    ]
    return lib


def _compact(lib, ring, tag_mask, scale=1.0, offset=0, max_out=1 << 22):
    out = np.zeros(max_out, dtype=REC_DTYPE)
    n_out = ctypes.c_uint32(0)
    rc = lib.sofa_ring_compact(
        ring,
        ctypes.c_uint64(tag_mask),
        ctypes.c_double(scale),
        ctypes.c_longlong(offset),
        out.ctypes.data_as(ctypes.c_void_p),
        ctypes.c_uint32(max_out),
        ctypes.byref(n_out),
    )
    assert rc == 0
    return out[: n_out.value]


def test_ring_produce_and_head(lib):
    ring = ctypes.c_void_p()
    assert lib.sofa_ring_create(0, 1 << 20, ctypes.byref(ring)) == 0
    try:
        assert lib.sofa_ring_test_produce(ring, 100000, 5) == 0
        head = ctypes.c_ulonglong(0)
        assert lib.sofa_ring_head(ring, ctypes.byref(head)) == 0
        assert head.value == 100000
    finally:
        lib.sofa_ring_destroy(ring)


def test_compaction_matches_cpu_reference(lib):
    n, n_tags = 200000, 7
    ring = ctypes.c_void_p()
    assert lib.sofa_ring_create(0, 1 << 20, ctypes.byref(ring)) == 0
    try:
        assert lib.sofa_ring_test_produce(ring, n, n_tags) == 0
        # keep tag classes 2 and 4 only
        mask = (1 << 2) | (1 << 4)
        got = _compact(lib, ring, mask)
        # CPU reference: producer pushes tag = 1 + (i % n_tags), arg = 3i+1
        i = np.arange(n, dtype=np.uint64)
        tags = (1 + (i % n_tags)).astype(np.uint32)
        keep = ((mask >> (tags & 63)) & 1).astype(bool)
        expected_args = np.sort((3 * i + 1)[keep])
        assert len(got) == keep.sum()
        assert np.array_equal(np.sort(got["arg"]), expected_args)
        # tags of kept records are only 2 and 4
        assert set(np.unique(got["tag"])) <= {2, 4}
        # t_end - t_start preserved (100 + i%7 at scale 1)
        deltas = got["t_end"] - got["t_start"]
        assert deltas.min() >= 100 and deltas.max() <= 106
    finally:
        lib.sofa_ring_destroy(ring)


def test_compaction_clock_transform(lib):
    """tick->ns fused transform: t' = t*scale + offset."""
    n = 50000
    ring = ctypes.c_void_p()
    assert lib.sofa_ring_create(0, 1 << 19, ctypes.byref(ring)) == 0
    try:
        assert lib.sofa_ring_test_produce(ring, n, 3) == 0
        raw = _compact(lib, ring, (1 << 1) | (1 << 2) | (1 << 3), scale=1.0, offset=0)
        scaled = _compact(lib, ring, (1 << 1) | (1 << 2) | (1 << 3), scale=10.0, offset=12345)
        assert len(raw) == len(scaled) == n
        r = np.sort(raw, order="arg")
        sc = np.sort(scaled, order="arg")
        expect = (r["t_start"].astype(np.float64) * 10.0 + 12345).astype(np.uint64)
        # double precision exact for these magnitudes? allow 1-ulp slack
        assert np.max(np.abs(sc["t_start"].astype(np.int64) - expect.astype(np.int64))) <= 1
    finally:
        lib.sofa_ring_destroy(ring)


def test_ring_wrap(lib):
    """Overfilling the ring keeps only the newest `capacity` records."""
    cap = 1 << 12
    n = cap * 3
    ring = ctypes.c_void_p()
    assert lib.sofa_ring_create(0, cap, ctypes.byref(ring)) == 0
    try:
        assert lib.sofa_ring_test_produce(ring, n, 1) == 0
        head = ctypes.c_ulonglong(0)
        lib.sofa_ring_head(ring, ctypes.byref(head))
        assert head.value == n
        got = _compact(lib, ring, 1 << 1)
        assert len(got) == cap  # only capacity survive
    finally:
        lib.sofa_ring_destroy(ring)
