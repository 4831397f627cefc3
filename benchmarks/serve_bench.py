"""Serving throughput: continuous batching vs serial per-request decode
on Llama-3-8B (bf16, 1x MI355X).  Decode is weight-read-bound, so slot
occupancy converts ~linearly into aggregate tokens/s."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from scaffold.config import load_config
from scaffold.generate import Generator
from scaffold.models import build_model
from scaffold.serve import ContinuousBatcher

N_REQ, PROMPT, NEW = 8, 128, 64

cfg = load_config(None, ["model.name=llama3_8b", "model.max_seq_len=4096"])
torch.manual_seed(0)
model = build_model(cfg).to("cuda", torch.bfloat16).eval()
prompts = [torch.randint(0, cfg.model.vocab_size, (PROMPT,)).tolist()
           for _ in range(N_REQ)]

# serial: one request at a time through the plain KV-cached decode
gen = Generator(model)
gen.generate(torch.tensor([prompts[0]], device="cuda"), 4)  # warmup
torch.cuda.synchronize()
t0 = time.perf_counter()
for p in prompts:
    gen.generate(torch.tensor([p], device="cuda"), NEW)
torch.cuda.synchronize()
dt_serial = time.perf_counter() - t0
print(f"serial: {N_REQ * NEW / dt_serial:.0f} tok/s "
      f"({dt_serial:.2f}s for {N_REQ}x{NEW})", flush=True)

# continuous: all requests share the slot pool
cb = ContinuousBatcher(model, max_slots=8)
try:
    t0 = time.perf_counter()
    reqs = [cb.submit(p, NEW) for p in prompts]
    for r in reqs:
        assert r.event.wait(timeout=600) and r.error is None
    dt_cont = time.perf_counter() - t0
finally:
    cb.close()
print(f"continuous (8 slots): {N_REQ * NEW / dt_cont:.0f} tok/s "
      f"({dt_cont:.2f}s, {dt_serial / dt_cont:.2f}x serial)", flush=True)
