"""Decode throughput (serving): tokens/s for KV-cached autoregressive
generation, GPT-2-small and a trimmed Llama config."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from scaffold.config import load_config
from scaffold.models import build_model
from scaffold.generate import Generator

def bench(name, overrides, B, prompt, new):
    cfg = load_config(None, overrides)
    model = build_model(cfg).to("cuda", torch.bfloat16).eval()
    gen = Generator(model)
    x = torch.randint(0, cfg.model.vocab_size, (B, prompt), device="cuda")
    # warmup
    gen.generate(x, max_new_tokens=8)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = gen.generate(x, max_new_tokens=new)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    tps = B * new / dt
    print(f"{name}: B={B} prompt={prompt} new={new}: {dt*1e3:.0f} ms, "
          f"{tps:.0f} tokens/s ({dt/new*1e3:.2f} ms/step)", flush=True)
    gen.generate_graphed(x, max_new_tokens=8)  # capture warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = gen.generate_graphed(x, max_new_tokens=new)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    tps = B * new / dt
    print(f"{name} GRAPHED: {dt*1e3:.0f} ms, {tps:.0f} tokens/s "
          f"({dt/new*1e3:.2f} ms/step)", flush=True)

bench("gpt2-small", [
    "model.name=gpt2", "model.vocab_size=50257", "model.n_layers=12",
    "model.n_heads=12", "model.d_model=768", "model.max_seq_len=1024",
], B=8, prompt=128, new=128)
bench("llama3-8b", ["model.name=llama3_8b", "model.max_seq_len=4096"],
      B=4, prompt=256, new=64)
