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

def sweep():
    """Batch sweep: serving-batching scaling (decode is weight-read-
    bound, so tokens/s should scale ~linearly in B until the batch
    GEMMs leave the skinny-M regime)."""
    cfg = load_config(None, ["model.name=llama3_8b", "model.max_seq_len=4096"])
    model = build_model(cfg).to("cuda", torch.bfloat16).eval()
    gen = Generator(model)
    for B in (1, 2, 4, 8):
        x = torch.randint(0, cfg.model.vocab_size, (B, 128), device="cuda")
        gen.generate(x, max_new_tokens=4)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        gen.generate(x, max_new_tokens=32)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        print(f"llama3-8b decode B={B}: {B*32/dt:.0f} tokens/s "
              f"({dt/32*1e3:.2f} ms/step)", flush=True)

if len(sys.argv) > 1 and sys.argv[1] == "sweep":
    sweep()
    sys.exit(0)

bench("gpt2-small", [
    "model.name=gpt2", "model.vocab_size=50257", "model.n_layers=12",
    "model.n_heads=12", "model.d_model=768", "model.max_seq_len=1024",
], B=8, prompt=128, new=128)
bench("llama3-8b", ["model.name=llama3_8b", "model.max_seq_len=4096"],
      B=4, prompt=256, new=64)
