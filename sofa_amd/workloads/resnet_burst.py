"""Short ResNet-50 training burst — the `sofa stat` demo workload
(BASELINE config 3).  Usage: python -m sofa_amd.workloads.resnet_burst
[--iters 6] [--batch 64]."""

import argparse

import torch
import torch.nn.functional as F

from .resnet import build_resnet50


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=6)
    ap.add_argument("--batch", type=int, default=64)
    args = ap.parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    m = build_resnet50(device=device, channels_last=device == "cuda")
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    x = torch.randn(args.batch, 3, 224, 224, device=device)
    if device == "cuda":
        x = x.to(memory_format=torch.channels_last)
    t = torch.randint(0, 1000, (args.batch,), device=device)
    for i in range(args.iters):
        with torch.autocast(device_type=device, dtype=torch.bfloat16):
            loss = F.cross_entropy(m(x), t)
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        print(f"iter {i}: loss {loss.item():.3f}")
    if device == "cuda":
        torch.cuda.synchronize()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
