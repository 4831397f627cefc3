"""K5 microbench: hand MFMA linear vs hipBLASLt (F.linear) at the
benched transformer shapes.  Run on a GPU box:

    python benchmarks/gemm_bench.py
"""

import sys
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from scaffold.ops.backend import require_ext


def time_fn(fn, iters=30, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    e.synchronize()
    return s.elapsed_time(e) / iters  # ms


def main():
    m = require_ext()
    dev = "cuda"
    print(f"{'shape':>24} {'epi':>6} {'hand ms':>9} {'blaslt ms':>10} {'hand TF':>8} {'ratio':>6}")
    shapes = [
        # (M, N, K, epi)  — GPT-2-small bs16 seq1024 and Llama-3-8B bs1 seq4096
        (16384, 2304, 768, "none"),    # gpt2 qkv
        (16384, 768, 768, "resid"),    # gpt2 attn proj + residual
        (16384, 3072, 768, "gelu"),    # gpt2 fc + bias + gelu
        (16384, 768, 3072, "resid"),   # gpt2 mlp proj + residual
        (4096, 6144, 4096, "none"),    # llama qkv-ish (q)
        (4096, 14336, 4096, "none"),   # llama gate/up
        (4096, 4096, 14336, "resid"),  # llama down + residual
    ]
    for M, N, K, epi in shapes:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        b = torch.randn(N, device=dev, dtype=torch.bfloat16)
        r = torch.randn(M, N, device=dev, dtype=torch.bfloat16)
        code = {"none": 0, "gelu": 1, "resid": 2}[epi]

        def hand():
            m.linear_fwd(x, w, b, code, r if epi == "resid" else None)

        if epi == "gelu":
            def ref():
                F.gelu(F.linear(x, w, b), approximate="tanh")
        elif epi == "resid":
            def ref():
                F.linear(x, w, b).add_(r)
        else:
            def ref():
                F.linear(x, w, b)

        th = time_fn(hand)
        tr = time_fn(ref)
        tf = 2 * M * N * K / (th * 1e-3) / 1e12
        print(f"{M}x{N}x{K:>6} {epi:>6} {th:9.3f} {tr:10.3f} {tf:8.0f} {tr/th:6.2f}")


if __name__ == "__main__":
    main()
