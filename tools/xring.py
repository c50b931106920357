#!/usr/bin/env python3
"""xring — sweep RCCL traffic across GPU-count / ring-order configurations.

Parity with reference tools/xring.py:34-72 (record: run `sofa stat` for
2..N GPUs with the ring-order hint applied; report: scrape measured traffic
into xring.csv), rebuilt for MI355X: the workload is a DDP all-reduce burst
over RCCL/xGMI and the scraped metrics come from features.csv +
xlink_traffic.csv.

Usage:
  python tools/xring.py record  [--max-gpus 8] [--logdir-base xringlog]
  python tools/xring.py report  [--logdir-base xringlog]
"""

import argparse
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SOFA = os.path.join(REPO, "bin", "sofa")

ALLREDUCE_SNIPPET = r"""
import os, torch, torch.distributed as dist
dist.init_process_group("nccl")
r = dist.get_rank(); torch.cuda.set_device(r)
x = torch.randn(64 << 20 >> 2, device="cuda")  # 64 MB fp32
for _ in range(50):
    dist.all_reduce(x)
torch.cuda.synchronize(); dist.destroy_process_group()
"""


def record(args):
    for n in range(2, args.max_gpus + 1):
        logdir = f"{args.logdir_base}-{n}"
        snip = os.path.join("/tmp", "xring_snippet.py")
        with open(snip, "w") as f:
            f.write(ALLREDUCE_SNIPPET)
        cmd = (
            f"python -m torch.distributed.run --nnodes=1 --nproc-per-node {n} "
            f"--master-addr 127.0.0.1 --master-port 29531 {snip}"
        )
        print(f"== recording {n} GPUs -> {logdir}")
        subprocess.run([sys.executable, SOFA, "stat", cmd, "--logdir", logdir], check=False)


def report(args):
    import pandas as pd

    rows = []
    n = 2
    while True:
        logdir = f"{args.logdir_base}-{n}"
        feat = os.path.join(logdir, "features.csv")
        if not os.path.isfile(feat):
            break
        f = pd.read_csv(feat)
        d = dict(zip(f["name"], f["value"]))
        row = {
            "n_gpus": n,
            "rccl_payload": d.get("rccl_payload", 0.0),
            "rccl_time": d.get("rccl_time", 0.0),
            "p2p_payload": d.get("p2p_payload", 0.0),
            "hot_link_bytes": d.get("rccl_hot_link_bytes", 0.0),
        }
        xl = os.path.join(logdir, "xlink_traffic.csv")
        if os.path.isfile(xl):
            x = pd.read_csv(xl)
            if len(x):
                row["hot_link_est_bw_GBps"] = x["est_bw_GBps"].max()
        rows.append(row)
        n += 1
    out = pd.DataFrame(rows)
    out.to_csv("xring.csv", index=False)
    print(out.to_string(index=False))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("verb", choices=["record", "report"])
    ap.add_argument("--max-gpus", type=int, default=8)
    ap.add_argument("--logdir-base", default="xringlog")
    args = ap.parse_args()
    (record if args.verb == "record" else report)(args)


if __name__ == "__main__":
    main()
