#!/usr/bin/env python3
"""train.py — CLI entrypoint (contract C1).

Usage:
    python train.py --config configs/resnet50.json [section.key=value ...]
    torchrun --nproc-per-node 8 train.py --config configs/resnet50.json

One process per GPU; reads RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* from the
launcher environment.
"""

from __future__ import annotations

import argparse
import sys

from scaffold.config import load_config
from scaffold.runtime import init_distributed, rank0_print
from scaffold.trainer import Trainer


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(description="MI355X-native training scaffold")
    ap.add_argument("--config", type=str, default=None, help="JSON/TOML config file")
    ap.add_argument("overrides", nargs="*", help="section.key=value overrides")
    args = ap.parse_args(argv)

    cfg = load_config(args.config, args.overrides)
    ctx = init_distributed(cfg)
    rank0_print(ctx, f"[scaffold] model={cfg.model.name} world={ctx.world_size} "
                     f"device={ctx.device} dtype={cfg.dtype} steps={cfg.steps}")
    trainer = Trainer(cfg, ctx=ctx)
    rank0_print(ctx, f"[scaffold] buckets(MB)={[round(b,1) for b in trainer.ddp.bucket_sizes_mb()]}")
    try:
        trainer.fit()
    finally:
        trainer.close()
    return 0


if __name__ == "__main__":
    sys.exit(main())
