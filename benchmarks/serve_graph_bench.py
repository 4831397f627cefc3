"""Continuous batching: eager vs hipGraph-replayed slot step
(llama3-8b bf16, 8 slots)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from scaffold.config import load_config
from scaffold.models import build_model
from scaffold.serve import ContinuousBatcher

N_REQ, PROMPT, NEW = 8, 128, 32

cfg = load_config(None, ["model.name=llama3_8b", "model.max_seq_len=4096"])
torch.manual_seed(0)
model = build_model(cfg).to("cuda", torch.bfloat16).eval()
prompts = [torch.randint(0, cfg.model.vocab_size, (PROMPT,)).tolist()
           for _ in range(N_REQ)]

results = {}
for use_graph in (False, True):
    cb = ContinuousBatcher(model, max_slots=8, use_graph=use_graph)
    try:
        # warm once (captures the graph on the graphed arm)
        w = [cb.submit(p, 4) for p in prompts[:8]]
        for r in w:
            assert r.event.wait(300) and r.error is None
        t0 = time.perf_counter()
        reqs = [cb.submit(p, NEW) for p in prompts]
        for r in reqs:
            assert r.event.wait(timeout=600) and r.error is None
        dt = time.perf_counter() - t0
        results[use_graph] = [r.result for r in reqs]
    finally:
        cb.close()
    tag = "graphed" if use_graph else "eager"
    print(f"continuous {tag}: {N_REQ * NEW / dt:.0f} tok/s ({dt:.2f}s)",
          flush=True)
assert results[False] == results[True], "graphed != eager"
print("parity OK", flush=True)
