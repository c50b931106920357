"""GPU e2e tests for the lite HSA-level collector (libsofahsalite):
dispatch/copy tracing via AQL packet interception + GPU-only pooled signals,
cross-checked against the rocprofiler-sdk collector on the same workload
(the repo-wide pattern: new path validated against an independent reference).
"""

import glob
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SOFA = os.path.join(REPO, "bin", "sofa")
LITE = os.path.join(REPO, "sofa_amd", "native", "lib", "libsofahsalite.so")
TRACER = os.path.join(REPO, "sofa_amd", "native", "lib", "libsofatracer.so")

TORCH_SNIPPET = (
    "import torch\n"
    "x = torch.randn(1024, 1024, device='cuda')\n"
    "for _ in range(10):\n"
    "    x = x @ x; x = x / x.norm()\n"
    "h = x.cpu()\n"
    "torch.cuda.synchronize()\n"
    "print('okay', float(h.sum()))\n"
)


def _run(cmd, extra_env, logdir):
    env = dict(os.environ)
    env["SOFA_LOGDIR"] = logdir
    env.update(extra_env)
    os.makedirs(logdir, exist_ok=True)
    return subprocess.run(cmd, env=env, capture_output=True, text=True, timeout=900)


def test_hsalite_traces_torch_kernels(tmp_path):
    """The lite collector alone must produce kernel + copy records with
    demanglable names and a sane clock pair."""
    logdir = str(tmp_path / "log")
    r = _run(
        [sys.executable, "-c", TORCH_SNIPPET],
        {"HSA_TOOLS_LIB": LITE},
        logdir,
    )
    assert "okay" in r.stdout, (r.stdout[-2000:], r.stderr[-2000:])
    sgts = glob.glob(os.path.join(logdir, "gputrace_*_lite.sgt"))
    assert sgts, "hsalite produced no SGT file — HSA_TOOLS_LIB not loaded"
    sys.path.insert(0, REPO)
    from sofa_amd.preprocess.sgt import parse_sgt

    s = parse_sgt(sgts[0])
    assert len(s.kernels) >= 10, "expected >=10 matmul kernel dispatches"
    # ROCclr's SDMA path calls hsa_amd_memory_async_copy_on_engine with a
    # NULL completion signal (measured); hsalite attaches its own to time it
    assert len(s.copies) >= 1, "no copy records (the .cpu() D2H)"
    assert s.kernel_names, "no kernel symbol names from executable_freeze"
    assert not any(nm.endswith(".kd") for nm in s.kernel_names.values())
    assert s.agents and any(a["device"] == 0 and a["type"] == 2 for a in s.agents)
    assert s.clocks
    k = s.kernels[0]
    assert k["end_ns"] > k["start_ns"]
    # GEMM on 1024^3 must take >1us and <1s
    dur = (s.kernels["end_ns"] - s.kernels["start_ns"]) * 1e-9
    assert (dur > 0).all() and (dur < 1.0).all()


def test_hsalite_matches_sdk_collector(tmp_path):
    """Numerics check: lite and SDK collectors trace the SAME deterministic
    workload in separate processes; the compute-kernel count must be
    identical and total kernel time within 25% (both ultimately read
    packet-processor timestamps)."""
    lite_dir = str(tmp_path / "lite")
    sdk_dir = str(tmp_path / "sdk")
    r1 = _run([sys.executable, "-c", TORCH_SNIPPET], {"HSA_TOOLS_LIB": LITE}, lite_dir)
    assert "okay" in r1.stdout, (r1.stdout[-2000:], r1.stderr[-2000:])
    r2 = _run(
        [sys.executable, "-c", TORCH_SNIPPET],
        {"ROCP_TOOL_LIBRARIES": TRACER, "SOFA_TRACE_HIP_API": "0",
         "SOFA_TRACE_RCCL": "0"},
        sdk_dir,
    )
    assert "okay" in r2.stdout, (r2.stdout[-2000:], r2.stderr[-2000:])
    sys.path.insert(0, REPO)
    from sofa_amd.preprocess.sgt import parse_sgt

    lite = parse_sgt(glob.glob(os.path.join(lite_dir, "gputrace_*_lite.sgt"))[0])
    sdk = parse_sgt(glob.glob(os.path.join(sdk_dir, "gputrace_*.sgt"))[0])
    assert len(sdk.kernels) > 0
    # identical deterministic launch sequence -> identical dispatch count
    assert abs(len(lite.kernels) - len(sdk.kernels)) <= 2, (
        len(lite.kernels),
        len(sdk.kernels),
    )
    t_lite = float((lite.kernels["end_ns"] - lite.kernels["start_ns"]).sum())
    t_sdk = float((sdk.kernels["end_ns"] - sdk.kernels["start_ns"]).sum())
    assert t_sdk > 0
    ratio = t_lite / t_sdk
    assert 0.75 < ratio < 1.33, f"total kernel time ratio {ratio}"


def test_sofa_stat_lite_mode(tmp_path):
    """Full pipeline with --gpu_tracer lite: Complete!! + named kernels in
    gputrace.csv."""
    logdir = str(tmp_path / "sofalog")
    code = (
        "import torch\n"
        "x = torch.randn(512, 512, device='cuda')\n"
        "for _ in range(20): x = torch.relu(x @ x)\n"
        "torch.cuda.synchronize(); print('okay')\n"
    )
    r = subprocess.run(
        [sys.executable, SOFA, "stat", f"{sys.executable} -c \"{code}\"",
         "--logdir", logdir, "--gpu_tracer", "lite"],
        capture_output=True, text=True, timeout=900,
    )
    assert "Complete!!" in r.stdout, (r.stdout[-3000:], r.stderr[-2000:])
    import pandas as pd

    df = pd.read_csv(os.path.join(logdir, "gputrace.csv"))
    kernels = df[df["copyKind"] == 0]
    assert len(kernels) >= 20
    # symbol names resolved (not the kernel_<object> fallback)
    assert (~kernels["name"].str.contains("kernel_")).any()
