"""Llama-3-8B-architecture decoder — the BASELINE config 5 stress workload
(high kernel-launch-rate training step profiled on 8 GPUs).

Own implementation, random init, synthetic tokens (no network for weights):
RMSNorm + RoPE + GQA (32 q heads / 8 kv heads) + SwiGLU, hidden 4096,
32 layers, intermediate 14336, vocab 128256 — sized for one MI355X's 288 GB
HBM3E under plain DDP (bf16 weights+grads+Adam fits with >100 GB headroom,
so no sharding complexity is needed on this hardware).

Run directly for a profiled step stress:
    sofa stat "python -m sofa_amd.workloads.llama --layers 8 --steps 3"
"""

from __future__ import annotations

import argparse

import torch
import torch.nn as nn
import torch.nn.functional as F


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x):
        dt = x.dtype
        x = x.float()
        x = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + self.eps)
        return (x * self.weight.float()).to(dt)


def precompute_rope(dim: int, max_seq: int, base: float = 500000.0, device="cpu"):
    inv = 1.0 / (base ** (torch.arange(0, dim, 2, device=device).float() / dim))
    t = torch.arange(max_seq, device=device).float()
    freqs = torch.outer(t, inv)
    return torch.cos(freqs), torch.sin(freqs)


def apply_rope(x, cos, sin):
    # x: (B, H, T, D)
    T = x.shape[-2]
    cos = cos[:T].to(x.dtype)
    sin = sin[:T].to(x.dtype)
    x1, x2 = x[..., ::2], x[..., 1::2]
    out = torch.empty_like(x)
    out[..., ::2] = x1 * cos - x2 * sin
    out[..., 1::2] = x2 * cos + x1 * sin
    return out


class Attention(nn.Module):
    def __init__(self, dim: int, n_heads: int, n_kv_heads: int):
        super().__init__()
        self.n_heads = n_heads
        self.n_kv = n_kv_heads
        self.hd = dim // n_heads
        self.wq = nn.Linear(dim, n_heads * self.hd, bias=False)
        self.wk = nn.Linear(dim, n_kv_heads * self.hd, bias=False)
        self.wv = nn.Linear(dim, n_kv_heads * self.hd, bias=False)
        self.wo = nn.Linear(n_heads * self.hd, dim, bias=False)

    def forward(self, x, cos, sin):
        B, T, C = x.shape
        q = self.wq(x).view(B, T, self.n_heads, self.hd).transpose(1, 2)
        k = self.wk(x).view(B, T, self.n_kv, self.hd).transpose(1, 2)
        v = self.wv(x).view(B, T, self.n_kv, self.hd).transpose(1, 2)
        q = apply_rope(q, cos, sin)
        k = apply_rope(k, cos, sin)
        # GQA: SDPA broadcasts kv heads with enable_gqa-free repeat
        k = k.repeat_interleave(self.n_heads // self.n_kv, dim=1)
        v = v.repeat_interleave(self.n_heads // self.n_kv, dim=1)
        o = F.scaled_dot_product_attention(q, k, v, is_causal=True)
        return self.wo(o.transpose(1, 2).reshape(B, T, C))


class MLP(nn.Module):
    def __init__(self, dim: int, hidden: int):
        super().__init__()
        self.w1 = nn.Linear(dim, hidden, bias=False)  # gate
        self.w3 = nn.Linear(dim, hidden, bias=False)  # up
        self.w2 = nn.Linear(hidden, dim, bias=False)  # down

    def forward(self, x):
        return self.w2(F.silu(self.w1(x)) * self.w3(x))


class Block(nn.Module):
    def __init__(self, dim, n_heads, n_kv, hidden):
        super().__init__()
        self.attn_norm = RMSNorm(dim)
        self.attn = Attention(dim, n_heads, n_kv)
        self.mlp_norm = RMSNorm(dim)
        self.mlp = MLP(dim, hidden)

    def forward(self, x, cos, sin):
        x = x + self.attn(self.attn_norm(x), cos, sin)
        return x + self.mlp(self.mlp_norm(x))


class Llama(nn.Module):
    def __init__(
        self,
        vocab: int = 128256,
        dim: int = 4096,
        n_layers: int = 32,
        n_heads: int = 32,
        n_kv_heads: int = 8,
        hidden: int = 14336,
        max_seq: int = 8192,
    ):
        super().__init__()
        self.embed = nn.Embedding(vocab, dim)
        self.blocks = nn.ModuleList(
            [Block(dim, n_heads, n_kv_heads, hidden) for _ in range(n_layers)]
        )
        self.norm = RMSNorm(dim)
        self.lm_head = nn.Linear(dim, vocab, bias=False)
        cos, sin = precompute_rope(dim // n_heads, max_seq)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def forward(self, tokens):
        x = self.embed(tokens)
        for b in self.blocks:
            x = b(x, self.rope_cos, self.rope_sin)
        return self.lm_head(self.norm(x))


def build_llama8b(device="cuda", n_layers: int = 32) -> Llama:
    m = Llama(n_layers=n_layers)
    return m.to(device)


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--layers", type=int, default=32)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--batch", type=int, default=2)
    ap.add_argument("--seq", type=int, default=4096)
    ap.add_argument("--device", default="cuda")
    args = ap.parse_args()

    torch.manual_seed(0)
    device = args.device if torch.cuda.is_available() or args.device == "cpu" else "cpu"
    model = build_llama8b(device=device, n_layers=args.layers)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-4)
    tokens = torch.randint(0, 128256, (args.batch, args.seq), device=device)
    target = torch.randint(0, 128256, (args.batch, args.seq), device=device)
    for i in range(args.steps):
        with torch.autocast(device_type="cuda" if device != "cpu" else "cpu", dtype=torch.bfloat16):
            logits = model(tokens)
            loss = F.cross_entropy(logits.view(-1, logits.shape[-1]).float(), target.view(-1))
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        print(f"step {i}: loss {loss.item():.4f}")
    if device != "cpu":
        torch.cuda.synchronize()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
