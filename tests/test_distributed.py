"""Multi-process CPU tests (gloo, world_size 2): the DDP/collective path of
bench.py and the cluster merge — runnable with no GPU."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_ddp_gloo_2proc(tmp_path):
    """bench.py under torch.distributed.run with 2 CPU ranks: the rank
    aggregation (MAX times, SUM events) and the JSON contract must hold."""
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29517",
            os.path.join(REPO, "bench.py"),
            "--gpus", "2", "--steps", "1", "--warmup", "0", "--batch", "2",
            "--no-profile",
        ],
        capture_output=True, text=True, timeout=900, env=env, cwd=REPO,
    )
    assert r.returncode == 0, r.stderr[-3000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert line, r.stdout
    d = json.loads(line[-1])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 4
    assert d["ms_per_step"] > 0
    assert d["scaling"] == "weak"
    assert d["dtype"] == "bf16"
    assert d["data"] == "synthetic"


def test_cluster_analyze_two_nodes(tmp_path):
    """Per-node logdirs <base>-<ip>/ merged into cluster_report.csv
    (reference bin/sofa:358-367 convention; arity bug not replicated)."""
    import pandas as pd

    sys.path.insert(0, REPO)
    from sofa_amd.analyze.main import cluster_analyze
    from sofa_amd.config import SofaConfig

    base = str(tmp_path / "clog")
    node_cfgs = {}
    for ip in ("10.0.0.1", "10.0.0.2"):
        d = f"{base}-{ip}/"
        os.makedirs(d)
        pd.DataFrame(
            {"name": ["elapsed_time", "gpu_time", "cpu_mean_busy"], "value": [12.5, 8.0, 55.0]}
        ).to_csv(os.path.join(d, "features.csv"), index=False)
        node_cfgs[ip] = SofaConfig(logdir=d)
    cfg = SofaConfig(logdir=base)
    cluster_analyze(cfg, node_cfgs)
    out = pd.read_csv(os.path.join(base, "cluster_report.csv"))
    assert len(out) == 2
    assert set(out["node"]) == {"10.0.0.1", "10.0.0.2"}
    assert (out["elapsed"] == 12.5).all()
