"""Collective layer used by the DDP engine.

A thin seam over torch.distributed (RCCL via the "nccl" backend on ROCm,
gloo on CPU) that the tests can replace with a recording fake
(fake_pg.RecordingComm).  Exposes generic collectives + sub-group
creation so TP/CP/EP strategies can compose later (SURVEY §5.7 seam b).

On ROCm, ProcessGroupNCCL enqueues each collective on RCCL's internal
comm stream after ordering it against the current compute stream, so an
``async_op=True`` all-reduce already runs concurrently with subsequent
backward kernels — the C4 overlap. ``work.wait()`` makes the compute
stream wait, never the host.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


class Comm:
    """Real collectives over the default (or a given) process group."""

    def __init__(self, group: Optional[dist.ProcessGroup] = None):
        self.group = group

    @property
    def is_distributed(self) -> bool:
        return dist.is_available() and dist.is_initialized()

    @property
    def world_size(self) -> int:
        return dist.get_world_size(self.group) if self.is_distributed else 1

    @property
    def rank(self) -> int:
        return dist.get_rank(self.group) if self.is_distributed else 0

    def all_reduce(self, t: torch.Tensor, async_op: bool = False):
        if not self.is_distributed:
            return None
        return dist.all_reduce(t, op=dist.ReduceOp.SUM, group=self.group, async_op=async_op)

    def broadcast(self, t: torch.Tensor, src: int = 0, async_op: bool = False):
        if not self.is_distributed:
            return None
        return dist.broadcast(t, src=src, group=self.group, async_op=async_op)

    def all_gather(self, out: List[torch.Tensor], t: torch.Tensor, async_op: bool = False):
        if not self.is_distributed:
            out[0].copy_(t)
            return None
        return dist.all_gather(out, t, group=self.group, async_op=async_op)

    def reduce_scatter(self, out: torch.Tensor, ins: List[torch.Tensor], async_op: bool = False):
        if not self.is_distributed:
            out.copy_(ins[0])
            return None
        return dist.reduce_scatter(out, ins, op=dist.ReduceOp.SUM, group=self.group, async_op=async_op)

    def all_to_all(self, out: List[torch.Tensor], ins: List[torch.Tensor], async_op: bool = False):
        if not self.is_distributed:
            out[0].copy_(ins[0])
            return None
        return dist.all_to_all(out, ins, group=self.group, async_op=async_op)

    def barrier(self):
        if self.is_distributed:
            dist.barrier(group=self.group)

    def new_group(self, ranks: List[int]) -> "Comm":
        if not self.is_distributed:
            return Comm()
        return Comm(dist.new_group(ranks))


class RcclBucketComm(Comm):
    """Bucket all-reduces through the direct-RCCL C++ core
    (csrc/comm/rccl_bucket.hip) on a dedicated comm stream; everything
    else (init broadcast, barriers, metrics) stays on torch.distributed.

    The communicator is bootstrapped from the torch process group: rank 0
    generates the ncclUniqueId, broadcast over the existing PG.  join()
    makes the compute stream wait on the comm stream (no host sync) —
    BucketedDDP calls it once at end-of-backward instead of per-work
    waits.  rccl_comm_time_ms() drains the measured per-bucket
    collective time for the §5.5 comm-exposed metric.
    """

    def __init__(self, device: torch.device, group=None):
        super().__init__(group)
        if not self.is_distributed:
            raise RuntimeError("RcclBucketComm needs torch.distributed")
        from ..ops.backend import require_ext

        self._ext = require_ext()
        idt = (self._ext.rccl_unique_id() if self.rank == 0
               else torch.zeros(128, dtype=torch.uint8))
        # broadcast the bootstrap id over whatever backend the PG has
        if dist.get_backend(self.group) == "nccl":
            idt = idt.to(device)
            dist.broadcast(idt, src=0, group=self.group)
            idt = idt.cpu()
        else:
            dist.broadcast(idt, src=0, group=self.group)
        self._handle = self._ext.rccl_init(self.rank, self.world_size, idt)

    def all_reduce_bucket(self, t: torch.Tensor) -> None:
        """Async SUM all-reduce on the comm stream (returns immediately)."""
        self._ext.rccl_all_reduce(self._handle, t)

    def join(self) -> None:
        """Compute stream waits for all outstanding bucket reduces."""
        self._ext.rccl_join(self._handle)

    def comm_time_ms(self) -> float:
        return self._ext.rccl_comm_time_ms(self._handle)


def make_comm(cfg, ctx) -> Comm:
    """Comm for the DDP engine: direct-RCCL on GPU under an
    initialized "nccl" (=RCCL) process group, plain torch.distributed
    otherwise (gloo/CPU tests, single-process runs)."""
    mode = getattr(cfg.dist, "comm", "auto")
    if (mode in ("auto", "rccl") and ctx.on_gpu and ctx.initialized
            and dist.get_backend() == "nccl"):
        try:
            return RcclBucketComm(ctx.device)
        except Exception:
            if mode == "rccl":
                raise
    return Comm()
