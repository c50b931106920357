"""Workload model tests (CPU: shapes/grads; GPU: traced stress step)."""

import os
import subprocess
import sys

import pytest
import torch

from sofa_amd.workloads.llama import Llama, build_llama8b
from sofa_amd.workloads.resnet import ResNet50


def test_resnet50_shapes_and_params():
    m = ResNet50()
    n_params = sum(p.numel() for p in m.parameters())
    # canonical ResNet-50 is ~25.5M params
    assert 25e6 < n_params < 26e6, n_params
    x = torch.randn(2, 3, 224, 224)
    y = m(x)
    assert y.shape == (2, 1000)
    y.square().mean().backward()
    assert m.conv1.weight.grad is not None


def test_llama_tiny_forward_backward():
    m = Llama(vocab=1000, dim=128, n_layers=2, n_heads=8, n_kv_heads=2, hidden=256, max_seq=64)
    t = torch.randint(0, 1000, (2, 32))
    logits = m(t)
    assert logits.shape == (2, 32, 1000)
    logits.float().square().mean().backward()
    assert m.blocks[0].attn.wq.weight.grad is not None


def test_llama8b_param_count():
    # count without materializing: meta device
    with torch.device("meta"):
        m = Llama()
    n = sum(p.numel() for p in m.parameters())
    assert 7.5e9 < n < 8.5e9, n  # Llama-3-8B class


@pytest.mark.gpu
def test_llama_stress_traced(tmp_path):
    """BASELINE config 5 (reduced layers): high-launch-rate step traced."""
    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    logdir = str(tmp_path / "log")
    os.makedirs(logdir)
    env = dict(os.environ)
    env["ROCP_TOOL_LIBRARIES"] = os.path.join(REPO, "sofa_amd", "native", "lib", "libsofatracer.so")
    env["SOFA_LOGDIR"] = logdir
    r = subprocess.run(
        [sys.executable, "-m", "sofa_amd.workloads.llama", "--layers", "4", "--steps", "2",
         "--batch", "1", "--seq", "2048"],
        env=env, capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert "step 1" in r.stdout, r.stderr[-2000:]
    import glob

    sys.path.insert(0, REPO)
    from sofa_amd.preprocess.sgt import parse_sgt

    sgts = glob.glob(os.path.join(logdir, "gputrace_*.sgt"))
    assert sgts
    s = parse_sgt(sgts[0])
    assert len(s.kernels) > 200, f"stress step traced only {len(s.kernels)} kernels"
    assert s.dropped == 0, f"collector dropped {s.dropped} records under stress"
