"""Run ONLY the hand 256^2 GEMM in a loop (for rocprofv3 PMC runs)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from scaffold.ops.backend import require_ext

m = require_ext()
M, N, K = 4096, 14336, 4096
x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.02
b = torch.randn(N, device="cuda", dtype=torch.bfloat16)
for _ in range(10):
    m.linear_fwd(x, w, b, 0, None)
torch.cuda.synchronize()
print("done")
