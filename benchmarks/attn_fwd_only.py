"""Run ONLY the attention forward kernel in a loop (for rocprofv3 PMC
runs, where per-kernel counters need a minimal kernel stream)."""

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from scaffold import ops

which = sys.argv[1] if len(sys.argv) > 1 else "fwd"
B, H, Hkv, S, D = 4, 32, 8, 4096, 128  # llama3-8b shape
q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)
if which == "fwd":
    for _ in range(10):
        o = ops.attention(q, k, v, True)
else:
    q.requires_grad_(True); k.requires_grad_(True); v.requires_grad_(True)
    do = torch.randn_like(q)
    for _ in range(5):
        o = ops.attention(q, k, v, True)
        o.backward(do)
        q.grad = k.grad = v.grad = None
torch.cuda.synchronize()
print("done", which)
