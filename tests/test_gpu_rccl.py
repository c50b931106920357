"""GPU e2e: RCCL collective tracing through the full pipeline (world_size 1 —
RCCL emits the same API surface; 8-GPU runs are driver territory)."""

import glob
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SOFA = os.path.join(REPO, "bin", "sofa")


def test_rccl_allreduce_traced(tmp_path):
    logdir = str(tmp_path / "log")
    snippet = tmp_path / "ar.py"
    snippet.write_text(
        "import os, torch, torch.distributed as dist\n"
        "os.environ.setdefault('MASTER_ADDR', '127.0.0.1')\n"
        "os.environ.setdefault('MASTER_PORT', '29533')\n"
        "os.environ.setdefault('RANK', '0')\n"
        "os.environ.setdefault('WORLD_SIZE', '1')\n"
        "dist.init_process_group('nccl')\n"
        "torch.cuda.set_device(0)\n"
        "x = torch.randn(1 << 20, device='cuda')\n"
        "for _ in range(5):\n"
        "    dist.all_reduce(x)\n"
        "torch.cuda.synchronize()\n"
        "dist.destroy_process_group()\n"
        "print('ar-done')\n"
    )
    r = subprocess.run(
        [sys.executable, SOFA, "stat", f"{sys.executable} {snippet}", "--logdir", logdir,
         "--gpu_tracer", "sdk"],
        capture_output=True, text=True, timeout=900,
    )
    assert "Complete!!" in r.stdout, (r.stdout[-2000:], r.stderr[-2000:])
    rccl_csv = os.path.join(logdir, "rccltrace.csv")
    assert os.path.isfile(rccl_csv), "no rccltrace.csv — RCCL API tracing failed"
    import pandas as pd

    df = pd.read_csv(rccl_csv)
    ar = df[df["name"].str.contains("ncclAllReduce")]
    assert len(ar) == 5, df["name"].tolist()[:10]
    # payload = 1M floats * 4 B
    assert (ar["payload"] == (1 << 20) * 4).all()
    assert (df["copyKind"] == 16).all()
    # analyzer picked it up
    feats = pd.read_csv(os.path.join(logdir, "features.csv"))
    d = dict(zip(feats["name"], feats["value"]))
    assert d.get("rccl_payload", 0) > 0
