"""Distributed runtime: process-group init, device binding, seeds, streams.

SURVEY.md §3.4: one OS process per GPU (torchrun-style launch); backend
"nccl" resolves to RCCL on ROCm, "gloo" is the CPU/test backend (C10).
"""

from __future__ import annotations

import datetime
import os
import random
from dataclasses import dataclass

import numpy as np
import torch
import torch.distributed as dist

from .config import TrainConfig


@dataclass
class DistContext:
    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    backend: str = "none"
    device: torch.device = torch.device("cpu")
    initialized: bool = False

    @property
    def is_main(self) -> bool:
        return self.rank == 0

    @property
    def on_gpu(self) -> bool:
        return self.device.type == "cuda"


def pick_backend(cfg: TrainConfig) -> str:
    if cfg.dist.backend != "auto":
        return cfg.dist.backend
    return "nccl" if torch.cuda.is_available() else "gloo"


def pick_device(cfg: TrainConfig, local_rank: int) -> torch.device:
    if cfg.device == "cpu":
        return torch.device("cpu")
    if cfg.device == "auto":
        if torch.cuda.is_available():
            return torch.device("cuda", local_rank)
        return torch.device("cpu")
    if cfg.device.startswith("cuda"):
        return torch.device("cuda", local_rank)
    return torch.device(cfg.device)


def init_distributed(cfg: TrainConfig) -> DistContext:
    """Initialize torch.distributed from launcher env vars.

    Reads RANK/WORLD_SIZE/LOCAL_RANK/MASTER_ADDR/MASTER_PORT.  Falls back
    to single-process mode (no process group) when WORLD_SIZE is absent
    or 1 and no group is needed.
    """
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    device = pick_device(cfg, local_rank)
    if device.type == "cuda":
        torch.cuda.set_device(device)
        # RCCL channel floor for the 7 xGMI links per GPU.
        os.environ.setdefault("NCCL_MIN_NCHANNELS", str(cfg.dist.min_nchannels))

    backend = pick_backend(cfg)
    initialized = False
    if dist.is_initialized():
        # adopt an externally-initialized process group (tests, embedders)
        rank = dist.get_rank()
        world = dist.get_world_size()
        backend = dist.get_backend()
        initialized = True
    elif world > 1 or "MASTER_ADDR" in os.environ:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world,
            timeout=datetime.timedelta(seconds=cfg.dist.timeout_s),
        )
        initialized = True
        # fail-fast sanity collective
        t = torch.ones(1, device=device if backend == "nccl" else "cpu")
        dist.all_reduce(t)
        if int(t.item()) != world:
            raise RuntimeError(
                f"sanity all_reduce returned {t.item()}, expected {world}"
            )
    set_seed(cfg.seed, rank)
    return DistContext(
        rank=rank,
        world_size=world,
        local_rank=local_rank,
        backend=backend if initialized else "none",
        device=device,
        initialized=initialized,
    )


def set_seed(seed: int, rank: int = 0) -> None:
    """Deterministic per-rank seeding (model init is rank-synced separately)."""
    s = seed * 1000003 + rank
    random.seed(s)
    np.random.seed(s % (2**32))
    torch.manual_seed(s)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(s)


def teardown(ctx: DistContext) -> None:
    if ctx.initialized and dist.is_initialized():
        dist.barrier()
        dist.destroy_process_group()


def barrier(ctx: DistContext) -> None:
    if ctx.initialized and dist.is_initialized():
        dist.barrier()


def rank0_print(ctx: DistContext, *args, **kwargs) -> None:
    if ctx.is_main:
        print(*args, **kwargs, flush=True)
