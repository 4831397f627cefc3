"""Run ONLY the fused BN kernels in a loop (rocprofv3 PMC probes)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from scaffold.ops.backend import require_ext

m = require_ext()
shapes = [(256, 64, 112, 112), (256, 256, 56, 56), (256, 512, 28, 28)]
for N, C, H, W in shapes:
    x = torch.randn(N, C, H, W, device="cuda", dtype=torch.bfloat16).to(
        memory_format=torch.channels_last)
    res = torch.randn_like(x)
    dy = torch.randn_like(x)
    w = torch.rand(C, device="cuda") + 0.5
    b = torch.randn(C, device="cuda")
    for _ in range(5):
        y, mean, rstd = m.bn_fwd(x, res, w, b, None, None, 0.1, 1e-5, True)
        m.bn_bwd(dy, x, res, w, b, mean, rstd, True)
torch.cuda.synchronize()
print("done")
