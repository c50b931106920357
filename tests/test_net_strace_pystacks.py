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


def test_parse_pystacks_multithread(tmp_path):
    """tid-tagged headers: per-thread duration diffing + thread-name prefix."""
    with open(os.path.join(tmp_path, "pystacks.txt.77"), "w") as f:
        # main thread at t, t+0.02; worker at t+0.01, t+0.03
        f.write("100.000000 111 MainThread\nmain (app.py:1);work (app.py:9)\n")
        f.write("100.010000 222 loader_0\nrun (dl.py:5);fetch (dl.py:44)\n")
        f.write("100.020000 111 MainThread\nmain (app.py:1);work (app.py:10)\n")
        f.write("100.030000 222 loader_0\nrun (dl.py:5);decode (dl.py:60)\n")
    df = parse_pystacks(str(tmp_path), None)
    assert set(df["tid"].unique()) == {111, 222}
    worker = df[df["tid"] == 222].sort_values("timestamp")
    assert len(worker) == 2
    # per-thread diff: 0.02 s between the two worker samples
    assert abs(worker["duration"].iloc[0] - 0.02) < 1e-9
    assert worker["name"].iloc[0].startswith("[loader_0] ")


def test_live_sampler_sees_all_threads(tmp_path):
    """End-to-end: inject the sampler into a child running two busy threads;
    both must appear in the output (round-1 gap: only main was sampled)."""
    import subprocess
    import sys

    inject = os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "sofa_amd",
        "pystacks_inject",
    )
    code = (
        "import threading, time\n"
        "def spin(n):\n"
        "    t0 = time.time()\n"
        "    while time.time() - t0 < 0.8: n = (n * 7 + 1) % 1000003\n"
        "th = threading.Thread(target=spin, args=(1,), name='worker_thread')\n"
        "th.start(); spin(2); th.join()\n"
    )
    env = dict(os.environ)
    env["PYTHONPATH"] = inject + os.pathsep + env.get("PYTHONPATH", "")
    env["SOFA_PYSTACKS_OUT"] = os.path.join(str(tmp_path), "pystacks.txt")
    env["SOFA_PYSTACKS_HZ"] = "100"
    subprocess.run([sys.executable, "-c", code], env=env, timeout=60, check=True)
    df = parse_pystacks(str(tmp_path), None)
    assert len(df) > 10
    names = df["name"].str.cat(sep="\n")
    assert "worker_thread" in names, "dataloader-style worker thread not sampled"
    assert "MainThread" in names
    assert df["tid"].nunique() >= 2
