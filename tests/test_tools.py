"""Unit tests for tools/ (xring report aggregation, doctor, etl bench)."""

import importlib.util
import json
import os
import subprocess
import sys

import pandas as pd
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _load(name, path):
    spec = importlib.util.spec_from_file_location(name, path)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def test_xring_report(tmp_path, monkeypatch):
    xring = _load("xring", os.path.join(REPO, "tools", "xring.py"))
    base = str(tmp_path / "xringlog")
    for n, payload in [(2, 1e6), (3, 2e6)]:
        d = f"{base}-{n}"
        os.makedirs(d)
        pd.DataFrame(
            {"name": ["rccl_payload", "rccl_time", "rccl_hot_link_bytes"],
             "value": [payload, 0.5, payload / n]}
        ).to_csv(os.path.join(d, "features.csv"), index=False)
        pd.DataFrame(
            {"src": [0], "dst": [1], "bytes": [payload], "time_s": [0.5],
             "est_bw_GBps": [payload / 0.5 / 1e9]}
        ).to_csv(os.path.join(d, "xlink_traffic.csv"), index=False)
    monkeypatch.chdir(tmp_path)

    class A:
        logdir_base = base

    xring.report(A())
    out = pd.read_csv(tmp_path / "xring.csv")
    assert len(out) == 2
    assert out["n_gpus"].tolist() == [2, 3]
    assert out["rccl_payload"].tolist() == [1e6, 2e6]


def test_doctor_runs():
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "doctor.py")],
        capture_output=True, text=True, timeout=300,
    )
    assert "ROCm (hipcc)" in r.stdout
    assert "native libsofatracer.so" in r.stdout


def test_etl_bench_small():
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "etl_bench.py"), "--n", "20000"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert r.returncode == 0, r.stderr[-1500:]
    assert "end-to-end ETL" in r.stdout


def test_sofa_edr_triggers(tmp_path, monkeypatch):
    """Event-driven recording: phase keyword in the watched log triggers a
    time-boxed `sofa record` (subprocess stubbed)."""
    edr = _load("sofa_edr", os.path.join(REPO, "tools", "sofa-edr.py"))
    log = tmp_path / "app.log"
    log.write_text("starting up\nphase: forward pass begins\n")
    calls = []

    def fake_run(cmd, check=False):
        calls.append(cmd)

        class R:
            returncode = 0

        return R()

    monkeypatch.setattr(edr.subprocess, "run", fake_run)
    monkeypatch.setattr(
        edr.sys, "argv",
        ["sofa-edr", "--watch", str(log), "--phases", "forward", "--duration", "5",
         "--logdir-base", str(tmp_path / "edr"), "--poll", "0.05"],
    )
    edr.main()
    assert len(calls) == 1
    assert "sleep 5" in " ".join(calls[0])
    assert any(str(tmp_path / "edr") + "-forward" in c for c in calls[0])


def test_lite_falls_back_when_rocprofiler_registered(monkeypatch):
    """A pre-set ROCP_TOOL_LIBRARIES (e.g. running under rocprofv3) would
    silently disable the lite collector — the recorder must fall back to the
    sdk path with a warning."""
    import os

    from sofa_amd.config import SofaConfig
    from sofa_amd.record.recorder import build_target_env

    monkeypatch.setenv("ROCP_TOOL_LIBRARIES", "/opt/rocm/lib/librocprofv3-whatever.so")
    cfg = SofaConfig(logdir="/tmp/x", gpu_tracer="lite")
    env = build_target_env(cfg)
    lite = "libsofahsalite.so"
    assert lite not in env.get("HSA_TOOLS_LIB", "")
    # sdk fallback appends the SDK collector after the user's tool
    assert "libsofatracer.so" in env.get("ROCP_TOOL_LIBRARIES", "")


def test_top_once_cpu():
    """`sofa top --once` renders one combined refresh without a GPU."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "bin", "sofa"), "top", "--once"],
        capture_output=True, text=True, timeout=120, cwd=repo,
    )
    assert r.returncode == 0, r.stderr[-1500:]
    assert "CPU" in r.stdout


def test_gpu_sample_env_plumbing():
    from sofa_amd.config import SofaConfig
    from sofa_amd.record.recorder import build_target_env

    cfg = SofaConfig(logdir="/tmp/x", gpu_tracer="lite", gpu_sample=16)
    env = build_target_env(cfg)
    if "libsofahsalite.so" in env.get("HSA_TOOLS_LIB", ""):
        assert env.get("SOFA_LITE_SAMPLE") == "16"
