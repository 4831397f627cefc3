"""Measure checkpoint save AND load throughput at GPT-2-small scale
(SURVEY §3.5 / VERDICT #7 'measured save AND load throughput')."""
import sys, os, time, shutil
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from scaffold.config import load_config
from scaffold.trainer import Trainer

out = sys.argv[1] if len(sys.argv) > 1 else "/tmp/ckpt_bench"
shutil.rmtree(out, ignore_errors=True)
cfg = load_config(None, [
    "model.name=gpt2", "data.kind=synthetic_tokens", "data.batch_size=4",
    "data.seq_len=512", "dtype=bf16", "log.interval=1000", f"ckpt.dir={out}",
])
t = Trainer(cfg)
it = iter(t.data)
t._train_step(it, 1)
torch.cuda.synchronize()
state_bytes = sum(p.numel() * p.element_size() for p in t.raw_model.parameters())
state_bytes += sum(m.numel() * 4 * 3 for m in t.optimizer.master)  # master+2 moments fp32
print(f"state ~{state_bytes/1e9:.2f} GB")

t.step = 1
t0 = time.perf_counter()
path = t.save_checkpoint()
dt = time.perf_counter() - t0
print(f"save: {dt:.2f}s  {state_bytes/dt/1e9:.2f} GB/s", flush=True)

import copy
cfg2 = copy.deepcopy(cfg)
cfg2.ckpt.resume = path
t0 = time.perf_counter()
t2 = Trainer(cfg2)  # includes model build; time load alone below
dt_total = time.perf_counter() - t0
from scaffold.ckpt.sharded import load_sharded
t0 = time.perf_counter()
load_sharded(path, model=t2.raw_model, optimizer=t2.optimizer, data=t2.data,
             ctx=t2.ctx, cfg_hash=t2.cfg_hash)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(f"load: {dt:.2f}s  {state_bytes/dt/1e9:.2f} GB/s (resume-incl-build {dt_total:.2f}s)", flush=True)
