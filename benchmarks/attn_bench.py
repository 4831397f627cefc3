#!/usr/bin/env python3
"""Attention kernel microbench: TFLOP/s at the contract model shapes.

Causal attention FLOP: ~2 * 2 * B*H*S^2/2*D (QK^T + PV) fwd;
bwd ~2.5x fwd.  Prints one line per shape for fwd and train (fwd+bwd).
"""

import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__file__) + "/..")
from scaffold import ops  # noqa: E402

SHAPES = [
    # (name, B, H, Hkv, S, D)
    ("gpt2", 16, 12, 12, 1024, 64),
    ("llama8b", 4, 32, 8, 4096, 128),
    ("guide", 16, 64, 8, 2048, 128),
]


def flops(B, H, S, D, causal=True):
    f = 4.0 * B * H * S * S * D
    return f / 2 if causal else f


def bench(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    for name, B, H, Hkv, S, D in SHAPES:
        q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)

        t = bench(lambda: ops.attention(q, k, v, True))
        tf = flops(B, H, S, D) / t / 1e12
        print(f"{name}: fwd {t*1e3:.2f} ms  {tf:.0f} TF/s (B{B} H{H} Hkv{Hkv} S{S} D{D})", flush=True)

        qg = q.clone().requires_grad_(True)
        kg = k.clone().requires_grad_(True)
        vg = v.clone().requires_grad_(True)
        do = torch.randn_like(q)

        def train_step():
            o = ops.attention(qg, kg, vg, True)
            o.backward(do)
            qg.grad = None
            kg.grad = None
            vg.grad = None

        t = bench(train_step, iters=5)
        tf = flops(B, H, S, D) * 3.5 / t / 1e12
        print(f"{name}: fwd+bwd {t*1e3:.2f} ms  {tf:.0f} TF/s-equiv", flush=True)

        # A/B: strided (B,S,H,D)->transpose views (the model's QKV layout)
        # vs the contiguous tensors above — isolates the stride cost in bwd
        qb = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
        kb = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
        vb = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
        qs = qb.transpose(1, 2).requires_grad_(True)
        ks = kb.transpose(1, 2).requires_grad_(True)
        vs = vb.transpose(1, 2).requires_grad_(True)

        def train_step_strided():
            o = ops.attention(qs, ks, vs, True)
            o.backward(do)
            qs.grad = None
            ks.grad = None
            vs.grad = None

        t2 = bench(train_step_strided, iters=5)
        print(f"{name}: fwd+bwd strided {t2*1e3:.2f} ms  (x{t2/t:.2f} vs contig)", flush=True)


if __name__ == "__main__":
    main()
