"""Bisect which stage of the ResNet train step rejects hipGraph capture
(r2: capture_error_mode=relaxed still fails with
hipErrorStreamCaptureUnsupported)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from scaffold.config import load_config
from scaffold.trainer import Trainer

cfg = load_config(None, [
    "model.name=resnet50", "data.kind=synthetic_image",
    "data.batch_size=32", "dtype=bf16", "log.interval=1000",
])
t = Trainer(cfg)
it = iter(t.data)
for _ in range(2):
    t._train_step(it, 1)
torch.cuda.synchronize()

def try_capture(name, fn):
    g = torch.cuda.CUDAGraph()
    try:
        with torch.cuda.graph(g, capture_error_mode="relaxed"):
            fn()
        torch.cuda.synchronize()
        print(f"{name}: OK", flush=True)
        return True
    except Exception as e:
        torch.cuda.synchronize()
        print(f"{name}: FAIL {str(e)[:120]}", flush=True)
        return False

x_static, y_static = t.data.next_batch()
try_capture("datagen", lambda: t.data.next_batch())
try_capture("fwd_static", lambda: t.model(x_static, y_static))
def fwd_bwd():
    t.ddp.zero_grad()
    loss = t.model(x_static, y_static)
    loss.backward()
    t.ddp.finish()
try_capture("fwd_bwd_static", fwd_bwd)
def opt_only():
    t.optimizer.step(grad_scale=1.0)
try_capture("opt", opt_only)

# second attempts: benchmark=False (heuristic algo mode) and a retry
# (first failure may have initialized whatever rejects capture)
torch.backends.cudnn.benchmark = False
for _ in range(2):
    t._train_step(it, 1)
torch.cuda.synchronize()
try_capture("fwd_bwd_heuristic", fwd_bwd)
try_capture("fwd_bwd_retry", fwd_bwd)
