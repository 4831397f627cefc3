"""Learning-rate schedules (optim.schedule): const / cosine / linear
with linear warmup.  Pure function of the step — checkpoint-resume safe
(the step count lives in the optimizer state)."""

from __future__ import annotations

import math

from .config import TrainConfig


def lr_at(cfg: TrainConfig, step: int, total_steps: int) -> float:
    o = cfg.optim
    base = o.lr
    if o.warmup_steps > 0 and step < o.warmup_steps:
        return base * (step + 1) / o.warmup_steps
    if o.schedule == "const":
        return base
    span = max(1, total_steps - o.warmup_steps)
    t = min(1.0, (step - o.warmup_steps) / span)
    floor = base * o.min_lr_ratio
    if o.schedule == "cosine":
        return floor + (base - floor) * 0.5 * (1.0 + math.cos(math.pi * t))
    if o.schedule == "linear":
        return floor + (base - floor) * (1.0 - t)
    raise ValueError(f"unknown optim.schedule {o.schedule!r}")
