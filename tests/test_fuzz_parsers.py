"""Property tests: the binary parsers must never crash on hostile input —
at worst raise ValueError (bad magic) or return empty results (partial
writes from killed collectors are a normal condition)."""

import struct

import pytest
from hypothesis import given, settings, strategies as st

from sofa_amd.config import SofaConfig
from sofa_amd.preprocess.scs import parse_scs
from sofa_amd.preprocess.sgt import parse_sgt
from sofa_amd.preprocess.strace import parse_sst
from sofa_amd.preprocess.net import parse_pktcap
from sofa_amd.preprocess.timebase import TimeBase

SCS_HDR = struct.pack("<IIQQQII", 0x31534353, 1, 10**18, 0, 0, 99, 8) + b"\0" * 32
SGT_HDR = struct.pack("<IIIIQQQ", 0x31544753, 1, 1, 0, 10**18, 0, 0) + b"\0" * 24
SST_HDR = struct.pack("<IIQQQ", 0x31545353, 1, 10**18, 0, 0)
SPC_HDR = struct.pack("<IIQQQ", 0x31435053, 1, 10**18, 0, 0)


def _write(tmp_path, name, data):
    p = tmp_path / name
    p.write_bytes(data)
    return str(p)


@settings(max_examples=40, deadline=None)
@given(st.binary(max_size=4096))
def test_scs_fuzz(tmp_path_factory, payload):
    tmp = tmp_path_factory.mktemp("fz")
    path = _write(tmp, "cpusamples.scs", SCS_HDR + payload)
    out = parse_scs(path)  # must not raise
    assert out.sample_freq == 99


@settings(max_examples=40, deadline=None)
@given(st.binary(max_size=4096))
def test_sgt_fuzz(tmp_path_factory, payload):
    tmp = tmp_path_factory.mktemp("fz")
    path = _write(tmp, "gputrace_1.sgt", SGT_HDR + payload)
    out = parse_sgt(path)  # must not raise
    assert out.pid == 1


@settings(max_examples=40, deadline=None)
@given(st.binary(max_size=4096))
def test_sst_fuzz(tmp_path_factory, payload):
    tmp = tmp_path_factory.mktemp("fz")
    _write(tmp, "strace.sst", SST_HDR + payload)
    cfg = SofaConfig(logdir=str(tmp))
    parse_sst(str(tmp), None, cfg)  # must not raise


@settings(max_examples=40, deadline=None)
@given(st.binary(max_size=4096))
def test_pktcap_fuzz(tmp_path_factory, payload):
    tmp = tmp_path_factory.mktemp("fz")
    _write(tmp, "pktcap.bin", SPC_HDR + payload)
    cfg = SofaConfig(logdir=str(tmp))
    parse_pktcap(str(tmp), None, cfg)  # must not raise


def test_bad_magic_raises(tmp_path):
    p = tmp_path / "cpusamples.scs"
    p.write_bytes(b"\xde\xad\xbe\xef" + b"\0" * 100)
    with pytest.raises(ValueError):
        parse_scs(str(p))


@settings(max_examples=40, deadline=None)
@given(st.text(max_size=2048))
def test_blkio_fuzz(tmp_path_factory, payload):
    from sofa_amd.preprocess.blkio import parse_blkio

    tmp = tmp_path_factory.mktemp("fz")
    _write(tmp, "blktrace.txt", payload.encode("utf-8", "replace"))
    parse_blkio(str(tmp), None)  # must not raise


@settings(max_examples=40, deadline=None)
@given(st.text(max_size=2048))
def test_rccl_log_fuzz(tmp_path_factory, payload):
    from sofa_amd.preprocess.rccl_log import parse_rccl_log

    tmp = tmp_path_factory.mktemp("fz")
    _write(tmp, "rccl_debug.h.1", payload.encode("utf-8", "replace"))
    parse_rccl_log(str(tmp))  # must not raise


@settings(max_examples=30, deadline=None)
@given(st.text(alphabet="0123456789. -x\n", max_size=1024))
def test_xgmi_counters_fuzz(tmp_path_factory, payload):
    from sofa_amd.preprocess.sysmon import parse_xgmi_counters

    tmp = tmp_path_factory.mktemp("fz")
    _write(tmp, "xgmi_counters.txt", payload.encode())
    try:
        parse_xgmi_counters(str(tmp), None)
    except Exception as e:  # only pandas parse errors may surface... no:
        raise AssertionError(f"xgmi parser raised {e!r}")


@settings(max_examples=30, deadline=None)
@given(st.text(alphabet="0123456789. -\n", max_size=512))
def test_gpusmi_7col_fuzz(tmp_path_factory, payload):
    from sofa_amd.preprocess.sysmon import parse_gpusmi

    tmp = tmp_path_factory.mktemp("fz")
    _write(tmp, "gpusmi.txt", payload.encode())
    parse_gpusmi(str(tmp), None)  # must not raise
