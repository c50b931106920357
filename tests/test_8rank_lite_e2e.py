"""Dress rehearsal for the driver's 8-GPU run, on CPU: a synthetic 8-rank
LITE-mode logdir (per-rank lite SGT files with ncclDevKernel dispatches, RCCL
debug logs, xGMI topology + HW counters) through the REAL preprocess +
analyze pipeline.  Everything the 8-GPU DDP bench produces must flow to
xlink_traffic.csv / xgmi_counters.csv / features without hardware."""

import json
import os

import numpy as np
import pandas as pd

from sgt_synth import SgtWriter
from sofa_amd.analyze.main import sofa_analyze
from sofa_amd.config import SofaConfig
from sofa_amd.preprocess.main import sofa_preprocess

N = 8
COUNT = 1 << 20  # elements; bf16 -> 2 MiB per collective


def build_logdir(tmp_path) -> str:
    logdir = str(tmp_path)
    with open(os.path.join(logdir, "sofa_time.txt"), "w") as f:
        f.write("1000.0\n")
    with open(os.path.join(logdir, "timebase.json"), "w") as f:
        json.dump({"realtime_ns": int(1000e9), "monotonic_raw_ns": 0}, f)
    # topology: all-to-all xGMI
    links = [
        [{"hops": 0, "type": 0, "weight": 0} if i == j
         else {"hops": 1, "type": 2, "weight": 15}
         for j in range(N)] for i in range(N)
    ]
    with open(os.path.join(logdir, "xgmi_topo.txt"), "w") as f:
        json.dump({"n_gpus": N, "links": links}, f)
    # per-rank lite SGT: compute kernels + one ncclDevKernel per iteration,
    # kernel_id values in the 64-bit kernel_object range (lite specific)
    for rank in range(N):
        w = SgtWriter(pid=2000 + rank, realtime_ns=int(1000e9) + 10**6,
                      monotonic_raw_ns=10**6, rocp_ns=10**9)
        w.agent(handle=70 + rank, device=rank)
        kid_gemm = (0x7F00 << 32) | 0x1000
        kid_rccl = (0x7F00 << 32) | 0x2000
        w.kernel_name(kid_gemm, "Cijk_Ailk_Bjlk_bf16_gemm")
        w.kernel_name(kid_rccl, "ncclDevKernel_AllReduce_Sum_bf16_RING_LL")
        base = w.rocp_ns
        for it in range(4):
            t0 = base + it * 20_000_000
            w.kernel(t0, t0 + 5_000_000, kid_gemm, device=rank, tid=100 + rank)
            w.kernel(t0 + 6_000_000, t0 + 8_000_000, kid_rccl, device=rank,
                     tid=100 + rank)
        w.write(os.path.join(logdir, "gputrace_%d_lite.sgt" % (2000 + rank)))
        # RCCL debug-log channel (what NCCL_DEBUG=INFO COLL writes per rank)
        with open(os.path.join(logdir, "rccl_debug.node01.%d" % (2000 + rank)), "w") as f:
            for it in range(4):
                f.write(
                    "node01:%d:%d [%d] NCCL INFO AllReduce: opCount %x "
                    "sendbuff 0x7f0 recvbuff 0x7f0 count %d datatype 9 op 0 "
                    "root 0 comm 0xc0ffee [nranks=%d] stream 0x9 task 0 "
                    "globalrank %d\n"
                    % (2000 + rank, 2100 + rank, rank, it, COUNT, N, rank)
                )
    # measured xGMI counters: ring-successor links carry traffic
    with open(os.path.join(logdir, "xgmi_counters.txt"), "w") as f:
        for tick in range(3):
            ts = 1000.0 + tick * 0.1
            for dev in range(N):
                reads = [0] * 8
                writes = [0] * 8
                writes[(dev + 1) % 8] = tick * 1_500_000  # KB accumulated
                f.write("%.6f %d %s %s\n" % (
                    ts, dev, " ".join(map(str, reads)), " ".join(map(str, writes))))
    return logdir


def test_8rank_lite_logdir_end_to_end(tmp_path, capsys):
    logdir = build_logdir(tmp_path)
    cfg = SofaConfig(logdir=logdir)
    pre = sofa_preprocess(cfg)

    # all 8 ranks' kernels merged
    gput = pd.read_csv(os.path.join(logdir, "gputrace.csv"))
    assert gput["pid"].nunique() == N
    assert gput["deviceId"].nunique() == N
    # collective args recovered from the debug-log channel
    rccl = pd.read_csv(os.path.join(logdir, "rccltrace.csv"))
    assert len(rccl) == 4 * N
    assert (rccl["payload"] == COUNT * 2).all()
    assert set(rccl["deviceId"].unique()) == set(range(N))
    # measured per-link counters
    xc = pd.read_csv(os.path.join(logdir, "xgmi_counters.csv"))
    assert (xc["GBps"] > 10).all()

    sofa_analyze(cfg, pre)
    out = capsys.readouterr().out
    assert "Complete!!" in out

    # ring-model per-link attribution with KERNEL-span denominators
    xl = pd.read_csv(os.path.join(logdir, "xlink_traffic.csv"))
    assert len(xl) == N
    S = COUNT * 2
    assert np.allclose(xl["bytes"], 4 * 2 * (N - 1) / N * S)
    # each collective matched to its 2 ms ncclDevKernel (4 per rank -> 8 ms)
    assert np.allclose(xl["time_s"], 4 * 2e-3, rtol=0.05)
    feats = pd.read_csv(os.path.join(logdir, "features.csv"))
    fmap = dict(zip(feats["name"], feats["value"]))
    assert fmap.get("rccl_kernel_match_ratio") == 1.0
    assert fmap.get("xgmi_meas_max_GBps", 0) > 10
