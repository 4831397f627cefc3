#!/usr/bin/env python3
"""bench.py — headline benchmark (driver contract).

Measures the BASELINE.json metric: samples/sec (whole node), ResNet-50
bf16 DDP, synthetic ImageNet-shape batches, random-init weights.

    python bench.py --gpus N --steps K --warmup W
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

W untimed warmup steps, then EXACTLY K timed steps bracketed by
barrier + torch.cuda.synchronize() on both sides; elapsed = MAX over
ranks; rank 0 prints ONE JSON line.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist

from scaffold.config import load_config
from scaffold.runtime import init_distributed
from scaffold.trainer import Trainer


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--model", type=str, default="resnet50",
                    choices=["resnet50", "gpt2", "llama3_8b", "mlp"])
    ap.add_argument("--batch-size", type=int, default=0, help="per-rank; 0=config default")
    ap.add_argument("--bucket-mb", type=float, default=0.0, help="bucket sweep override")
    ap.add_argument("--config", type=str, default="")
    ap.add_argument("overrides", nargs="*", help="section.key=value overrides")
    args = ap.parse_args()

    cfg_file = args.config or {
        "resnet50": "configs/resnet50.json",
        "gpt2": "configs/gpt2_small.json",
        "llama3_8b": "configs/llama3_8b.json",
        "mlp": "configs/mlp_cpu.json",
    }[args.model]
    here = os.path.dirname(os.path.abspath(__file__))
    cfg = load_config(os.path.join(here, cfg_file), args.overrides)
    if args.batch_size:
        cfg.data.batch_size = args.batch_size
    if args.bucket_mb:
        cfg.dist.bucket_cap_mb = args.bucket_mb
    cfg.log.interval = 10**9  # silent during timing
    cfg.ckpt.interval = 0
    cfg.ckpt.save_final = False

    ctx = init_distributed(cfg)
    on_gpu = ctx.on_gpu
    t = Trainer(cfg, ctx=ctx)
    batch_iter = iter(t.data)
    accum = max(1, cfg.grad_accum)

    use_graph = False
    if t._graph_ok():
        try:
            t._capture_step(batch_iter)
            use_graph = True
        except Exception as e:  # noqa: BLE001 — e.g. non-capturable RNG op
            print(f"[bench] hipGraph capture failed ({e}); eager", flush=True)

    if use_graph:
        def one_step():
            t._train_step_graph()
    else:
        def one_step():
            t._train_step(batch_iter, accum)

    for _ in range(args.warmup):
        one_step()

    if ctx.initialized:
        dist.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if ctx.initialized:
        # MAX over ranks
        e = torch.tensor([elapsed], dtype=torch.float64,
                         device=ctx.device if ctx.backend == "nccl" else "cpu")
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())
        dist.barrier()

    n_gpus = ctx.world_size
    global_batch = cfg.data.batch_size * accum * n_gpus
    is_lm = args.model in ("gpt2", "llama3_8b")
    samples = global_batch * args.steps
    value = samples / elapsed
    unit = "samples/s"
    metric = "samples_per_sec_node"
    extra_cfg = {}
    if is_lm:
        # secondary configs also report tokens/s for readability
        extra_cfg["tokens_per_sec"] = round(value * cfg.data.seq_len, 1)

    if ctx.is_main:
        out = {
            "metric": metric,
            "value": round(value, 2),
            "unit": unit,
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # no published reference numbers
            "dtype": cfg.dtype,
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": global_batch,
                "seq_len": cfg.data.seq_len if is_lm else None,
                "image_size": cfg.data.image_size if args.model == "resnet50" else None,
                "parallelism": f"dp{n_gpus}",
                "bucket_cap_mb": cfg.dist.bucket_cap_mb,
                **extra_cfg,
            },
        }
        print(json.dumps(out), flush=True)
    t.close()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
