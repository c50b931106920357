#!/usr/bin/env python3
"""Profiling overhead on a transformer step (Llama-architecture, reduced
layers): complements bench.py's ResNet-50 number — transformer steps have
far fewer, longer kernels, so the per-dispatch interception cost amortizes
differently.

Run on a GPU box: python tools/overhead_llama.py [--layers 8 --steps 10]
"""

import argparse
import ctypes
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
TRACER = os.path.join(REPO, "sofa_amd", "native", "lib", "libsofatracer.so")


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--layers", type=int, default=8)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--batch", type=int, default=2)
    ap.add_argument("--seq", type=int, default=2048)
    ap.add_argument("--tracer", choices=["sdk", "lite"], default="lite")
    args = ap.parse_args()

    logdir = os.path.join(REPO, "gpurun_out", "llama_ovh")
    os.makedirs(logdir, exist_ok=True)
    from bench import setup_tracer_env, TracerCtl

    setup_tracer_env(logdir, args.tracer)

    import torch
    import torch.nn.functional as F

    from sofa_amd.workloads.llama import build_llama8b

    lib = TracerCtl(args.tracer)

    model = build_llama8b(device="cuda", n_layers=args.layers)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-4)
    tokens = torch.randint(0, 128256, (args.batch, args.seq), device="cuda")
    target = torch.randint(0, 128256, (args.batch, args.seq), device="cuda")

    def step():
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            logits = model(tokens)
            loss = F.cross_entropy(logits.view(-1, logits.shape[-1]).float(), target.view(-1))
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()

    def timed(k):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(k):
            step()
        torch.cuda.synchronize()
        return time.perf_counter() - t0

    for _ in range(4):
        step()
    torch.cuda.synchronize()

    # interleaved A/B like bench.py
    t_plain = t_prof = 0.0
    n0 = lib.event_count()
    per = max(args.steps // 2, 1)
    for _ in range(2):
        t_plain += timed(per)
        lib.start()
        t_prof += timed(per)
        lib.stop()
    n_ev = int(lib.event_count() - n0)

    steps = per * 2
    print(("llama-%dL b%d s%d [" + args.tracer + "]: plain %.1f ms/step, "
           "profiled %.1f ms/step, overhead %.2f%%, %d events (%.0f ev/s)")
          % (args.layers, args.batch, args.seq,
             t_plain / steps * 1e3, t_prof / steps * 1e3,
             100 * (t_prof - t_plain) / t_plain, n_ev, n_ev / t_prof))
    return 0


if __name__ == "__main__":
    sys.exit(main())
