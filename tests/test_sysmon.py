"""Poller TSV parsers — synthetic raw files, delta math verification."""

import os

import numpy as np
import pytest

from sofa_amd.preprocess import sysmon
from sofa_amd.preprocess.timebase import TimeBase

TB = TimeBase(time_base=100.0, realtime_ns=100_000_000_000, monotonic_raw_ns=0)


def test_parse_mpstat(tmp_path):
    # core 0: 50% busy in tick 2 (50 jiffies busy of 100), core 1 idle
    lines = [
        "100.0 0 1000 0 500 8500 0 0 0 0",
        "100.0 1 100 0 100 9800 0 0 0 0",
        "101.0 0 1040 0 510 8550 0 0 0 0",   # +40 usr +10 sys +50 idle
        "101.0 1 100 0 100 9900 0 0 0 0",    # all idle
    ]
    (tmp_path / "mpstat.txt").write_text("\n".join(lines) + "\n")
    tdf, mp_csv, usr_sys = sysmon.parse_mpstat(str(tmp_path), TB)
    assert len(tdf) == 2
    core0 = mp_csv[mp_csv["core"] == 0].iloc[0]
    assert np.isclose(core0["usr_r"], 40.0)
    assert np.isclose(core0["sys_r"], 10.0)
    assert np.isclose(core0["busy_r"], 50.0)
    core1 = mp_csv[mp_csv["core"] == 1].iloc[0]
    assert np.isclose(core1["busy_r"], 0.0)
    # timeline = ts - time_base
    assert np.isclose(tdf["timestamp"].iloc[0], 1.0)


def test_parse_diskstat(tmp_path):
    # nvme0: 100 reads, 2048 sectors (1 MiB) in 1 s
    lines = [
        "100.0 nvme0 1000 20480 500 2000 40960 800 0",
        "101.0 nvme0 1100 22528 510 2100 43008 810 0",
        "100.0 zero 0 0 0 0 0 0 0",
        "101.0 zero 0 0 0 0 0 0 0",
    ]
    (tmp_path / "diskstat.txt").write_text("\n".join(lines) + "\n")
    tdf, vec = sysmon.parse_diskstat(str(tmp_path), TB)
    assert len(vec) == 1  # all-zero device dropped
    row = vec.iloc[0]
    assert row["dev"] == "nvme0"
    assert np.isclose(row["r_iops"], 100.0)
    assert np.isclose(row["read_Bps"], 2048 * 512)
    assert np.isclose(row["r_await_ms"], 10 / 100)


def test_parse_netstat(tmp_path):
    lines = [
        "100.0 eth0 1000000 1000 2000000 2000",
        "101.0 eth0 2000000 1500 4000000 2500",
        "100.0 lo 5 5 5 5",
        "101.0 lo 9 9 9 9",
    ]
    (tmp_path / "netstat.txt").write_text("\n".join(lines) + "\n")
    tdf, bw = sysmon.parse_netstat(str(tmp_path), TB)
    assert len(bw) == 1  # lo excluded
    assert np.isclose(bw["rx_Bps"].iloc[0], 1e6)
    assert np.isclose(bw["tx_Bps"].iloc[0], 2e6)


def test_parse_vmstat(tmp_path):
    lines = [
        "100.0 1000 2000 0 0 50000 30000 2 0",
        "101.0 1100 2400 0 0 51000 30500 3 1",
    ]
    (tmp_path / "vmstat.txt").write_text("\n".join(lines) + "\n")
    tdf, vm = sysmon.parse_vmstat(str(tmp_path), TB)
    assert np.isclose(vm["ctxt_r"].iloc[0], 1000.0)
    assert np.isclose(vm["pgpgin_r"].iloc[0], 100.0)


def test_parse_gpusmi(tmp_path):
    lines = [
        "100.0 0 80 40 1000000 500.0",
        "100.0 1 20 10 2000000 400.0",
        "100.1 0 90 45 1000000 510.0",
    ]
    (tmp_path / "gpusmi.txt").write_text("\n".join(lines) + "\n")
    sm, mem, csv = sysmon.parse_gpusmi(str(tmp_path), TB)
    assert len(sm) == 3
    assert sm[sm["deviceId"] == 0]["duration"].max() == 90
    assert mem[mem["deviceId"] == 1]["duration"].iloc[0] == 10


def test_missing_files_return_empty(tmp_path):
    tdf, mp, us = sysmon.parse_mpstat(str(tmp_path), TB)
    assert len(tdf) == 0
    tdf, vec = sysmon.parse_diskstat(str(tmp_path), TB)
    assert len(tdf) == 0
    sm, mem, csv = sysmon.parse_gpusmi(str(tmp_path), TB)
    assert len(sm) == 0


def test_parse_gpusmi_mm_column(tmp_path):
    """7-column gpusmi (with MM/media engine busy) -> event-2 series; the
    nvsmi dmon enc/dec analog (reference bin/sofa_preprocess.py:1097-1183)."""
    with open(os.path.join(tmp_path, "gpusmi.txt"), "w") as f:
        f.write("100.0 0 80 40 1000000 500.0 25\n")
        f.write("100.1 0 85 45 1000000 505.0 30\n")
    sm, mem, csv = sysmon.parse_gpusmi(str(tmp_path), None)
    assert len(sm) == 2
    mm = mem[mem["event"] == 2.0]
    assert len(mm) == 2
    assert list(mm["duration"]) == [25.0, 30.0]
    assert "gpu0_mm:25%" in mm["name"].iloc[0]


def test_parse_gpusmi_legacy_6col(tmp_path):
    with open(os.path.join(tmp_path, "gpusmi.txt"), "w") as f:
        f.write("100.0 0 80 40 1000000 500.0\n")
    sm, mem, csv = sysmon.parse_gpusmi(str(tmp_path), None)
    assert len(sm) == 1
    assert (mem["event"] == 2.0).sum() == 0  # no MM series


def test_parse_xgmi_counters(tmp_path):
    """HW accumulator deltas -> per-link GB/s (measured ground truth for the
    analytic ring model)."""
    r0 = [0] * 8
    r1 = [0] * 8
    r1[3] = 15_000_000  # link 3: 15 GB read over 0.1 s -> 150 GB/s
    w1 = [0] * 8
    w1[3] = 1_000_000
    with open(os.path.join(tmp_path, "xgmi_counters.txt"), "w") as f:
        f.write("100.0 0 %s %s\n" % (" ".join(map(str, r0)), " ".join(map(str, r0))))
        f.write("100.1 0 %s %s\n" % (" ".join(map(str, r1)), " ".join(map(str, w1))))
    trace, csv = sysmon.parse_xgmi_counters(str(tmp_path), None)
    assert len(trace) == 2  # read + write rows on link 3
    rd = csv[(csv["kind"] == "read") & (csv["link"] == 3)]
    assert len(rd) == 1
    assert abs(rd["GBps"].iloc[0] - 150.0) < 1.0
    wr = csv[csv["kind"] == "write"]
    assert abs(wr["GBps"].iloc[0] - 10.0) < 0.2


def test_xgmi_measured_profile_features(tmp_path):
    import pandas as pd

    from sofa_amd.analyze import profiles

    pd.DataFrame(
        {"ts": [1.0, 1.1], "dev": [0, 0], "link": [3, 3],
         "kind": ["read", "read"], "GBps": [120.0, 140.0]}
    ).to_csv(os.path.join(tmp_path, "xgmi_counters.csv"), index=False)
    feats = []
    profiles.xgmi_measured_profile(str(tmp_path), feats)
    d = dict(feats)
    assert d["xgmi_meas_max_GBps"] == 140.0
    assert d["xgmi_meas_links_active"] == 1.0
