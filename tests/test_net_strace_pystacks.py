"""Parser tests: pktcap binary, syscall trace binary, pystacks text."""

import os
import struct

import numpy as np
import pytest

from sofa_amd.config import SofaConfig
from sofa_amd.preprocess.net import PKT_DTYPE, pack_ip_base1000, parse_pktcap
from sofa_amd.preprocess.pystacks import parse_pystacks
from sofa_amd.preprocess.strace import SYS_DTYPE, parse_sst, syscall_names
from sofa_amd.preprocess.timebase import TimeBase

TB = TimeBase(time_base=100.0, realtime_ns=100_000_000_000, monotonic_raw_ns=0)


def test_pack_ip_base1000():
    ip = np.array([0x0A000001], dtype=np.uint64)  # 10.0.0.1
    assert pack_ip_base1000(ip)[0] == 10_000_000_001


def test_parse_pktcap(tmp_path):
    cfg = SofaConfig(logdir=str(tmp_path))
    hdr = struct.pack("<IIQQQ", 0x31435053, 1, 100_000_000_000, 0, 0)
    recs = np.zeros(3, dtype=PKT_DTYPE)
    recs["time_ns"] = [1_000_000_000, 2_000_000_000, 3_000_000_000]
    recs["src_ip"] = 0x0A000001
    recs["dst_ip"] = 0x0A000002
    recs["sport"] = 1234
    recs["dport"] = 80
    recs["len"] = [100, 200, 1500]
    recs["proto"] = 6
    with open(os.path.join(tmp_path, "pktcap.bin"), "wb") as f:
        f.write(hdr + recs.tobytes())
    df = parse_pktcap(str(tmp_path), TB, cfg)
    assert len(df) == 3
    assert df["pkt_src"].iloc[0] == 10_000_000_001
    assert df["pkt_dst"].iloc[0] == 10_000_000_002
    assert "tcp" in df["name"].iloc[0]
    assert "10.0.0.1:1234_to_10.0.0.2:80" in df["name"].iloc[0]
    assert np.isclose(df["timestamp"].iloc[0], 1.0)


def test_parse_sst(tmp_path):
    cfg = SofaConfig(logdir=str(tmp_path), strace_min_time=1e-5)
    hdr = struct.pack("<IIQQQ", 0x31545353, 1, 100_000_000_000, 0, 0)
    recs = np.zeros(3, dtype=SYS_DTYPE)
    recs["t_enter_ns"] = [1_000_000_000, 2_000_000_000, 3_000_000_000]
    recs["duration_ns"] = [500_000, 5_000, 300_000]  # middle one < min_time
    recs["tid"] = 42
    recs["sysno"] = [0, 1, 202]  # read, write, futex (noisy -> dropped)
    recs["ret"] = [4096, 128, 0]
    with open(os.path.join(tmp_path, "strace.sst"), "wb") as f:
        f.write(hdr + recs.tobytes())
    df = parse_sst(str(tmp_path), TB, cfg)
    assert len(df) == 1  # write dropped (too short), futex dropped (noisy)
    assert df["name"].iloc[0].startswith("read(")
    assert df["payload"].iloc[0] == 4096


def test_syscall_names_table():
    names = syscall_names()
    assert names[0] == "read"
    assert names[59] == "execve"


def test_parse_pystacks(tmp_path):
    with open(os.path.join(tmp_path, "pystacks.txt.123"), "w") as f:
        f.write("101.0\nmain (app.py:10);train (app.py:50)\n")
        f.write("101.1\nmain (app.py:10);loss (app.py:80)\n")
        f.write("101.2\nmain (app.py:10);wait (thr.py:5)\n")  # idle -> dropped
    df = parse_pystacks(str(tmp_path), TB)
    assert len(df) == 2
    assert (df["pid"] == 123).all()
    assert "<br>" in df["name"].iloc[0]
    assert np.isclose(df["timestamp"].iloc[0], 1.0)
