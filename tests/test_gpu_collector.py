"""GPU e2e tests: the rocprofiler-sdk collector traces a real HIP workload;
the SGT trace parses into kernels/copies with symbol names; the full
`sofa stat` pipeline produces gputrace.csv + the Complete!! sentinel
(BASELINE configs 2-3)."""

import glob
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SOFA = os.path.join(REPO, "bin", "sofa")
TRACER = os.path.join(REPO, "sofa_amd", "native", "lib", "libsofatracer.so")
BANDWIDTH = os.path.join(REPO, "sofa_amd", "native", "bin", "sofa-bandwidth")


def _run_traced(cmd, logdir, extra_env=None):
    env = dict(os.environ)
    env["ROCP_TOOL_LIBRARIES"] = TRACER
    env["SOFA_LOGDIR"] = logdir
    env.update(extra_env or {})
    os.makedirs(logdir, exist_ok=True)
    return subprocess.run(cmd, env=env, capture_output=True, text=True, timeout=900)


def test_collector_traces_torch_kernels(tmp_path):
    """A torch matmul on cuda:0 must produce kernel + copy records with names."""
    logdir = str(tmp_path / "log")
    code = (
        "import torch\n"
        "x = torch.randn(1024, 1024, device='cuda')\n"
        "y = x @ x\n"
        "h = y.cpu()\n"
        "torch.cuda.synchronize()\n"
        "print('okay', float(h.sum()))\n"
    )
    r = _run_traced([sys.executable, "-c", code], logdir)
    assert "okay" in r.stdout, r.stderr[-2000:]
    sgts = glob.glob(os.path.join(logdir, "gputrace_*.sgt"))
    assert sgts, "collector produced no SGT file — native tracer not loaded"

    sys.path.insert(0, REPO)
    from sofa_amd.preprocess.sgt import parse_sgt

    s = parse_sgt(sgts[0])
    assert len(s.kernels) >= 1, "no kernel dispatch records"
    assert len(s.copies) >= 1, "no memory copy records (the .cpu() D2H)"
    assert s.kernel_names, "no kernel symbol names captured"
    assert s.agents, "no agent records"
    assert any(a["device"] == 0 and a["type"] == 2 for a in s.agents)
    assert s.clocks, "no clock correlation records"
    # timestamps sane: kernels within the clock pair's vicinity
    rt, mono, rocp = s.clocks[0]
    assert rocp > 0
    k = s.kernels[0]
    assert k["end_ns"] > k["start_ns"]


def test_sofa_stat_bandwidth_workload(tmp_path):
    """BASELINE config 2: profile the HIP bandwidth workload end-to-end."""
    logdir = str(tmp_path / "sofalog")
    r = subprocess.run(
        [
            sys.executable,
            SOFA,
            "stat",
            f"{BANDWIDTH} 0",
            "--logdir",
            logdir,
        ],
        capture_output=True,
        text=True,
        timeout=900,
    )
    assert "Complete!!" in r.stdout, (r.stdout[-3000:], r.stderr[-2000:])
    assert os.path.isfile(os.path.join(logdir, "gputrace.csv"))
    import pandas as pd

    df = pd.read_csv(os.path.join(logdir, "gputrace.csv"))
    kinds = set(df["copyKind"].unique())
    assert 1 in kinds, "no H2D copies traced"
    assert 2 in kinds, "no D2H copies traced"
    assert 0 in kinds, "no kernels traced (copy_kernel)"
    kernels = df[df["copyKind"] == 0]
    assert kernels["name"].str.contains("copy_kernel").any()
    # bandwidths should be physically plausible on MI355X (H2D < 128 GB/s)
    h2d = df[(df["copyKind"] == 1) & (df["payload"] > (1 << 20))]
    assert (h2d["bandwidth"] < 800e9).all()
    assert (h2d["bandwidth"] > 1e8).any()


def test_sofa_stat_torch_resnet_like(tmp_path):
    """BASELINE config 3 (reduced): a small conv net fwd+bwd traced."""
    logdir = str(tmp_path / "sofalog")
    code = (
        "import torch, torch.nn as nn\n"
        "m = nn.Sequential(nn.Conv2d(3, 64, 7, 2, 3), nn.BatchNorm2d(64), nn.ReLU(),\n"
        "                  nn.Conv2d(64, 64, 3, 1, 1), nn.AdaptiveAvgPool2d(1),\n"
        "                  nn.Flatten(), nn.Linear(64, 10)).cuda()\n"
        "opt = torch.optim.SGD(m.parameters(), lr=0.01)\n"
        "x = torch.randn(16, 3, 64, 64, device='cuda')\n"
        "t = torch.randint(0, 10, (16,), device='cuda')\n"
        "for _ in range(3):\n"
        "    loss = nn.functional.cross_entropy(m(x), t)\n"
        "    opt.zero_grad(); loss.backward(); opt.step()\n"
        "torch.cuda.synchronize(); print('okay')\n"
    )
    r = subprocess.run(
        [sys.executable, SOFA, "stat", f"{sys.executable} -c \"{code}\"", "--logdir", logdir],
        capture_output=True,
        text=True,
        timeout=900,
    )
    assert "Complete!!" in r.stdout, (r.stdout[-3000:], r.stderr[-2000:])
    import pandas as pd

    df = pd.read_csv(os.path.join(logdir, "gputrace.csv"))
    kernels = df[df["copyKind"] == 0]
    assert len(kernels) > 50, "expected many kernels from 3 training steps"
    # MIOpen/rocBLAS kernels appear with demangled-ish names
    assert kernels["name"].str.len().max() > 10


def test_pc_sampling_hotspots(tmp_path):
    """GPU PC sampling (--pc_sampling, experimental SDK API): samples land
    in pcsamples.csv attributed to kernels.  Skips when the driver/stack
    rejects host-trap sampling (the collector prints its verdict)."""
    logdir = str(tmp_path / "log")
    code = (
        "import torch\n"
        "x = torch.randn(4096, 4096, device='cuda')\n"
        "for _ in range(60):\n"
        "    x = (x @ x)\n"
        "    x = x / x.norm()\n"
        "torch.cuda.synchronize(); print('okay')\n"
    )
    r = subprocess.run(
        [sys.executable, SOFA, "stat", f"{sys.executable} -c \"{code}\"",
         "--logdir", logdir, "--pc_sampling"],
        capture_output=True, text=True, timeout=900,
    )
    assert "Complete!!" in r.stdout, (r.stdout[-3000:], r.stderr[-2000:])
    if "pc sampling active" not in r.stderr and "pc sampling active" not in r.stdout:
        import pytest as _pytest

        _pytest.skip("host-trap PC sampling not available on this stack: "
                     + (r.stderr.splitlines()[-1] if r.stderr else "?"))
    csv_path = os.path.join(logdir, "pcsamples.csv")
    assert os.path.isfile(csv_path), "pc sampling active but no pcsamples.csv"
    import pandas as pd

    df = pd.read_csv(csv_path)
    assert len(df) > 10, "too few PC samples for a 60-matmul loop"
    # the GEMM must dominate the sample mass
    top = df.groupby("kernel").size().sort_values(ascending=False)
    assert any("Cijk" in str(k) or "gemm" in str(k).lower() for k in top.index[:2]), top.index[:3]
    assert df["active_lanes"].between(0, 64).all()
