"""SGT binary parser + GPU preprocess tests (GPU-free, synthetic traces)."""

import os

import numpy as np
import pytest

from sgt_synth import SgtWriter
from sofa_amd.preprocess.gpu import sgt_to_gputrace, sgt_to_hip_api_trace, sgt_to_rccltrace
from sofa_amd.preprocess.sgt import parse_sgt
from sofa_amd.preprocess.timebase import TimeBase


@pytest.fixture
def synth(tmp_path):
    w = SgtWriter(pid=77, realtime_ns=1_000_000_000_000, rocp_ns=100_000)
    w.clock(realtime_ns=1_000_000_000_000, mono=50_000, rocp=100_000)
    w.agent(handle=1, device=0)
    w.agent(handle=2, device=1)
    w.kernel_name(11, "_Z10add_kernelPfS_")
    w.kernel_name(12, "rccl_allreduce_kernel")
    w.opname(9, 2, "HOST_TO_DEVICE")
    w.opname(203, 1, "ncclAllReduce")
    # two kernels on gpu0, one on gpu1
    w.kernel(start=200_000, end=300_000, kid=11, device=0)
    w.kernel(start=300_000, end=350_000, kid=12, device=0)
    w.kernel(start=220_000, end=280_000, kid=11, device=1)
    # one H2D copy of 1 MB taking 100 us
    w.copy(start=400_000, end=500_000, op=2, nbytes=1 << 20, src=-1, dst=0)
    # one P2P copy gpu0 -> gpu1
    w.copy(start=500_000, end=520_000, op=4, nbytes=1 << 16, src=0, dst=1)
    # one allreduce of 1M bf16 elems
    w.rccl(start=600_000, end=700_000, op=1, count=1 << 20, elem_size=2)
    w.hip_api(start=100_000, end=110_000, op=5)
    path = tmp_path / "gputrace_77.sgt"
    w.write(str(path))
    return str(path)


def test_parse_sgt_counts(synth):
    s = parse_sgt(synth)
    assert s.pid == 77
    assert len(s.kernels) == 3
    assert len(s.copies) == 2
    assert len(s.rccl) == 1
    assert len(s.hip_api) == 1
    assert s.kernel_names[11] == "_Z10add_kernelPfS_"
    assert len(s.agents) == 2
    assert s.clocks == [(1_000_000_000_000, 50_000, 100_000)]


def test_sgt_to_gputrace_schema(synth):
    s = parse_sgt(synth)
    tb = TimeBase(time_base=1000.0, realtime_ns=1_000_000_000_000, monotonic_raw_ns=50_000)
    df = sgt_to_gputrace([s], tb)
    assert len(df) == 5  # 3 kernels + 2 copies
    kernels = df[df["copyKind"] == 0]
    assert len(kernels) == 3
    # demangled name with [gpuN] prefix
    assert any("add_kernel" in n for n in kernels["name"])
    assert any(n.startswith("[gpu1]") for n in kernels["name"])
    # kernel durations: (300000-200000) ns = 100 us
    assert np.isclose(kernels["duration"].max(), 100e-6)
    # timeline placement: rocp 200000 + (rt - rocp offset) -> ~0 s after base
    assert (df["timestamp"].abs() < 1.0).all()
    # copy kinds: H2D=1, P2P=10
    h2d = df[df["copyKind"] == 1]
    p2p = df[df["copyKind"] == 10]
    assert len(h2d) == 1 and len(p2p) == 1
    assert h2d["payload"].iloc[0] == 1 << 20
    # bandwidth = 1MB / 100us ~ 10.5 GB/s
    assert np.isclose(h2d["bandwidth"].iloc[0], (1 << 20) / 100e-6, rtol=1e-3)
    assert p2p["pkt_src"].iloc[0] == 0 and p2p["pkt_dst"].iloc[0] == 1


def test_sgt_to_rccltrace(synth):
    s = parse_sgt(synth)
    df = sgt_to_rccltrace([s], None)
    assert len(df) == 1
    assert df["copyKind"].iloc[0] == 16
    assert df["payload"].iloc[0] == (1 << 20) * 2
    assert "ncclAllReduce" in df["name"].iloc[0]


def test_sgt_to_hip_api(synth):
    s = parse_sgt(synth)
    df = sgt_to_hip_api_trace([s], None)
    assert len(df) == 1
    assert df["category"].iloc[0] == 1


def test_truncated_sgt_is_tolerated(synth, tmp_path):
    with open(synth, "rb") as f:
        buf = f.read()
    trunc = tmp_path / "gputrace_t.sgt"
    trunc.write_bytes(buf[: len(buf) - 37])  # cut mid-record
    s = parse_sgt(str(trunc))
    assert len(s.kernels) >= 1  # earlier records still parse


def test_markers_to_timeline(tmp_path):
    from sofa_amd.preprocess.gpu import sgt_to_markers

    w = SgtWriter(pid=5, realtime_ns=2_000_000_000_000, rocp_ns=0)
    w.clock(realtime_ns=2_000_000_000_000, mono=0, rocp=0)
    w.marker(500_000, "step_begin")
    w.marker(900_000, "step_end")
    path = tmp_path / "gputrace_5.sgt"
    w.write(str(path))
    s = parse_sgt(str(path))
    assert len(s.markers) == 2
    df = sgt_to_markers([s], None)
    assert len(df) == 2
    assert df["name"].tolist() == ["roctx:step_begin", "roctx:step_end"]
    assert abs(df["timestamp"].iloc[0] - 2000.0005) < 1e-6


def test_pcsample_records_roundtrip(tmp_path):
    """PC-sample records: synth write -> parse -> kernel-attributed frame."""
    from sgt_synth import SgtWriter
    from sofa_amd.preprocess.gpu import sgt_to_pcsamples
    from sofa_amd.preprocess.sgt import parse_sgt

    w = SgtWriter()
    w.kernel_name(5, "_Z8hot_loopv")
    w.kernel(w.rocp_ns, w.rocp_ns + 1000, 5, corr=77)
    for i in range(6):
        w.pcsample(w.rocp_ns + 100 * i, corr=77, offset=0x40 + 16 * (i % 2),
                   exec_mask=(1 << 32) - 1)
    path = os.path.join(str(tmp_path), "gputrace_1.sgt")
    w.write(path)
    s = parse_sgt(path)
    assert len(s.pcsamples) == 6
    df = sgt_to_pcsamples([s], None)
    assert len(df) == 6
    assert (df["kernel"] == "hot_loop()").all()
    assert (df["active_lanes"] == 32).all()  # half-divergent wave
    assert set(df["offset"]) == {0x40, 0x50}


def test_pc_hotspot_profile(tmp_path, capsys):
    import pandas as pd

    from sofa_amd.analyze.profiles import pc_hotspot_profile

    pd.DataFrame(
        {
            "timestamp": [0.1] * 10,
            "kernel": ["hot_kernel"] * 8 + ["cold_kernel"] * 2,
            "code_object_id": 1,
            "offset": [0x40] * 6 + [0x80] * 2 + [0x10] * 2,
            "active_lanes": [64] * 8 + [8] * 2,
            "dispatch_id": 0,
            "wave_in_group": 0,
            "pid": 1,
        }
    ).to_csv(os.path.join(str(tmp_path), "pcsamples.csv"), index=False)
    feats = []
    pc_hotspot_profile(str(tmp_path), feats)
    d = dict(feats)
    assert d["pcsamples_total"] == 10
    out = capsys.readouterr().out
    assert "hot_kernel" in out and "80.0%" in out
    assert "+0x40" in out
