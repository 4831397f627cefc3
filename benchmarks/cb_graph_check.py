import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from scaffold.config import load_config
from scaffold.models import build_model
from scaffold.serve import ContinuousBatcher

cfg = load_config(None, [
    "model.name=llama", "model.n_layers=2", "model.n_heads=4",
    "model.n_kv_heads=2", "model.d_model=128", "model.d_ff=256",
    "model.vocab_size=512", "model.max_seq_len=128",
])
torch.manual_seed(43)
m = build_model(cfg).to("cuda", torch.bfloat16).eval()

def run(use_graph):
    cb = ContinuousBatcher(m, max_slots=4, use_graph=use_graph)
    try:
        reqs = [cb.submit([i + 1, i + 2, i + 3], 12) for i in range(6)]
        for r in reqs:
            assert r.event.wait(timeout=120), "timeout"
            assert r.error is None, r.error
        return [r.result for r in reqs]
    finally:
        cb.close()

a = run(False)
b = run(True)
assert a == b, (a, b)
print("GRAPHED CONTINUOUS OK", a[0][:4])
