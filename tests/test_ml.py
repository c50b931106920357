"""HSG swarm clustering + diff tests (synthetic CPU traces)."""

import os

import numpy as np
import pandas as pd
import pytest

from sofa_amd.config import SofaConfig
from sofa_amd.ml.diff import sofa_swarm_diff
from sofa_amd.ml.fuzz import best_match, ratio
from sofa_amd.ml.hsg import hsg_cluster, swarms_to_traces
from sofa_amd.schema import new_trace_df


def test_fuzz_ratio():
    assert ratio("abc", "abc") == 100
    assert ratio("abc", "xyz") == 0
    assert 50 < ratio("matmul_kernel_a", "matmul_kernel_b") < 100
    m, s = best_match("gemm_fp16", ["conv", "gemm_fp32", "relu"])
    assert m == "gemm_fp32"


def _synth_cpu(funcs, n_per=50, base_event=14.0):
    n = len(funcs) * n_per
    df = new_trace_df(n)
    rng = np.random.default_rng(0)
    events, names, ts = [], [], []
    for i, f in enumerate(funcs):
        for j in range(n_per):
            events.append(base_event + i + rng.normal(0, 0.01))
            names.append(f)
            ts.append(j * 0.01)
    df["event"] = events
    df["name"] = names
    df["timestamp"] = ts
    df["duration"] = 0.01
    return df


def test_hsg_cluster_groups_by_event(tmp_path):
    funcs = ["funcA @ a.so", "funcB @ b.so", "funcC @ c.so"]
    df = _synth_cpu(funcs)
    out, captions = hsg_cluster(df, 3, str(tmp_path))
    assert len(captions) == 3
    assert set(captions) == set(funcs)
    # samples of one function land in one cluster
    for f in funcs:
        sel = out[out["name"] == f]
        assert sel["cluster_ID"].nunique() == 1
    assert os.path.isfile(os.path.join(tmp_path, "swarms_report.txt"))
    assert os.path.isfile(os.path.join(tmp_path, "auto_caption.csv"))
    traces = swarms_to_traces(out, captions)
    assert len(traces) == 3


def test_swarm_diff(tmp_path):
    base_dir = tmp_path / "base"
    match_dir = tmp_path / "match"
    base_dir.mkdir()
    match_dir.mkdir()
    df1 = _synth_cpu(["alpha @ x.so", "beta @ y.so"])
    df2 = _synth_cpu(["alpha @ x.so", "gamma @ z.so"])
    hsg_cluster(df1, 2, str(base_dir))
    hsg_cluster(df2, 2, str(match_dir))
    cfg = SofaConfig(logdir=str(tmp_path), base_logdir=str(base_dir), match_logdir=str(match_dir))
    out = sofa_swarm_diff(cfg)
    assert len(out) == 2
    alpha = out[out["caption"].str.contains("alpha")]
    assert len(alpha) == 1 and alpha["match_cluster"].iloc[0] >= 0
    assert os.path.isfile(os.path.join(str(tmp_path), "swarm_diff.csv"))


def test_diff_cli_verb(tmp_path):
    """`sofa diff` end-to-end through the CLI (with --skip_preprocess)."""
    import subprocess
    import sys

    base_dir = tmp_path / "base"
    match_dir = tmp_path / "match"
    base_dir.mkdir()
    match_dir.mkdir()
    hsg_cluster(_synth_cpu(["alpha @ x.so", "beta @ y.so"]), 2, str(base_dir))
    hsg_cluster(_synth_cpu(["alpha @ x.so", "beta @ y.so"]), 2, str(match_dir))
    sofa = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "bin", "sofa")
    r = subprocess.run(
        [
            sys.executable, sofa, "diff",
            "--base_logdir", str(base_dir),
            "--match_logdir", str(match_dir),
            "--logdir", str(tmp_path),
            "--skip_preprocess",
        ],
        capture_output=True, text=True, timeout=120,
    )
    assert r.returncode == 0, r.stderr
    assert "intersection rate" in r.stdout
    assert os.path.isfile(os.path.join(str(tmp_path), "swarm_diff.csv"))
