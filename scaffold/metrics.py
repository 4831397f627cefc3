"""Metrics/logging: rank-0 JSONL + stdout, samples/sec,
step-time breakdown with exposed-comm measured by CUDA events."""

from __future__ import annotations

import json
import time
from collections import deque
from typing import Dict, Optional

import torch


class MetricLogger:
    def __init__(self, ctx, cfg, window: int = 50):
        self.ctx = ctx
        self.cfg = cfg
        self.step_times = deque(maxlen=window)
        self.fh = None
        if ctx.is_main and cfg.log.metrics_file:
            self.fh = open(cfg.log.metrics_file, "a", buffering=1)

    def log_step(self, step: int, loss: float, samples: int, step_time_s: float,
                 extra: Optional[Dict] = None) -> None:
        self.step_times.append(step_time_s)
        if step % max(1, self.cfg.log.interval) != 0:
            return
        if not self.ctx.is_main:
            return
        avg_t = sum(self.step_times) / len(self.step_times)
        node_sps = samples * self.ctx.world_size / avg_t if avg_t > 0 else 0.0
        rec = {
            "ts": time.time(),
            "step": step,
            "loss": round(loss, 6),
            "ms_per_step": round(avg_t * 1000, 3),
            "samples_per_sec_node": round(node_sps, 2),
            "world_size": self.ctx.world_size,
        }
        if extra:
            rec.update(extra)
        if self.fh:
            self.fh.write(json.dumps(rec) + "\n")
        print(
            f"step {step:6d} | loss {loss:8.4f} | {avg_t*1000:8.2f} ms/step "
            f"| {node_sps:10.1f} samples/s (node)",
            flush=True,
        )

    def close(self) -> None:
        if self.fh:
            self.fh.close()


class CudaTimer:
    """Event-based GPU span timer; host-clock fallback on CPU."""

    def __init__(self, device: torch.device):
        self.on_gpu = device.type == "cuda"
        if self.on_gpu:
            self.start_ev = torch.cuda.Event(enable_timing=True)
            self.end_ev = torch.cuda.Event(enable_timing=True)
        self._t0 = 0.0

    def start(self):
        if self.on_gpu:
            self.start_ev.record()
        else:
            self._t0 = time.perf_counter()

    def stop_ms(self) -> float:
        if self.on_gpu:
            self.end_ev.record()
            self.end_ev.synchronize()
            return self.start_ev.elapsed_time(self.end_ev)
        return (time.perf_counter() - self._t0) * 1000
