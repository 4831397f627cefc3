#!/usr/bin/env python3
"""bench.py — headline benchmark (driver contract).

Measures the BASELINE.json metric: samples/sec (whole node), ResNet-50
bf16 DDP, synthetic ImageNet-shape batches, random-init weights.

    python bench.py --gpus N --steps K --warmup W
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

W untimed warmup steps, then EXACTLY K timed steps bracketed by
barrier + torch.cuda.synchronize() on both sides; elapsed = MAX over
ranks; rank 0 prints ONE JSON line.  Protocol per BASELINE.md: default
warmup >= 20, >= 100 measured steps, median (+p10/p90) per-step time
reported alongside the aggregate.

--sweep-buckets "4,8,16,25,50" reruns the timed loop per bucket size
(fresh Trainer each point, same process group) and emits one JSON line
per point — the §5.8 bucket table in a single launch.
"""

from __future__ import annotations

import argparse
import json
import os
import statistics
import time

import torch
import torch.distributed as dist

from scaffold.config import load_config
from scaffold.runtime import init_distributed
from scaffold.trainer import Trainer


def run_once(cfg, ctx, steps: int, warmup: int) -> dict:
    """One measured bench run: returns elapsed (max over ranks), per-step
    quantiles, comm evidence and peak HBM."""
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

    for _ in range(warmup):
        one_step()

    if ctx.initialized:
        dist.barrier()
    if on_gpu:
        torch.cuda.synchronize()
        torch.cuda.reset_peak_memory_stats(ctx.device)
        if hasattr(t.ddp.comm, "comm_time_ms"):
            t.ddp.comm.comm_time_ms()  # drain warmup's collective timers
        # per-step marks: stream-ordered events, queried after the run —
        # no host syncs inside the timed region
        marks = [torch.cuda.Event(enable_timing=True) for _ in range(steps + 1)]
    t0 = time.perf_counter()
    if on_gpu:
        marks[0].record()
    host_marks = [t0]
    for i in range(steps):
        one_step()
        if on_gpu:
            marks[i + 1].record()
        else:
            host_marks.append(time.perf_counter())
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

    if on_gpu:
        per_step = [marks[i].elapsed_time(marks[i + 1]) for i in range(steps)]
    else:
        per_step = [(host_marks[i + 1] - host_marks[i]) * 1000 for i in range(steps)]
    per_step.sort()
    q = lambda f: per_step[min(steps - 1, int(f * steps))]  # noqa: E731
    out = {
        "elapsed_s": elapsed,
        "ms_per_step_median": round(statistics.median(per_step), 3),
        "ms_per_step_p10": round(q(0.10), 3),
        "ms_per_step_p90": round(q(0.90), 3),
    }
    if hasattr(t.ddp.comm, "comm_time_ms"):
        out["comm_ms_per_step"] = round(t.ddp.comm.comm_time_ms() / steps, 3)
    if on_gpu:
        out["hbm_peak_gb"] = round(
            torch.cuda.max_memory_allocated(ctx.device) / (1 << 30), 2)
    out["graph"] = use_graph
    # free the model/optimizer before a subsequent sweep point
    del t, batch_iter
    if on_gpu:
        torch.cuda.empty_cache()
    return out


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=100)
    ap.add_argument("--warmup", type=int, default=20)
    ap.add_argument("--model", type=str, default="resnet50",
                    choices=["resnet50", "gpt2", "llama3_8b", "mlp"])
    ap.add_argument("--batch-size", type=int, default=0, help="per-rank; 0=config default")
    ap.add_argument("--bucket-mb", type=float, default=0.0, help="bucket override")
    ap.add_argument("--sweep-buckets", type=str, default="",
                    help="comma-separated bucket MiB list; one JSON line each")
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
    n_gpus = ctx.world_size
    accum = max(1, cfg.grad_accum)
    global_batch = cfg.data.batch_size * accum * n_gpus
    is_lm = args.model in ("gpt2", "llama3_8b")

    def emit(res: dict, bucket_mb: float, sweep: bool) -> None:
        if not ctx.is_main:
            return
        samples = global_batch * args.steps
        value = samples / res["elapsed_s"]
        extra_cfg = {}
        if is_lm:
            extra_cfg["tokens_per_sec"] = round(value * cfg.data.seq_len, 1)
        out = {
            "metric": "samples_per_sec_node",
            "value": round(value, 2),
            "unit": "samples/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(res["elapsed_s"] / args.steps * 1000, 3),
            "ms_per_step_median": res["ms_per_step_median"],
            "ms_per_step_p10": res["ms_per_step_p10"],
            "ms_per_step_p90": res["ms_per_step_p90"],
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
                "bucket_cap_mb": bucket_mb,
                **extra_cfg,
            },
        }
        for k in ("comm_ms_per_step", "hbm_peak_gb"):
            if k in res:
                out[k] = res[k]
        if sweep:
            out["sweep"] = True
        print(json.dumps(out), flush=True)

    if args.sweep_buckets:
        for mb in [float(s) for s in args.sweep_buckets.split(",") if s]:
            cfg.dist.bucket_cap_mb = mb
            res = run_once(cfg, ctx, args.steps, args.warmup)
            emit(res, mb, sweep=True)
    else:
        res = run_once(cfg, ctx, args.steps, args.warmup)
        emit(res, cfg.dist.bucket_cap_mb, sweep=False)

    if ctx.initialized:
        dist.barrier()
        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
