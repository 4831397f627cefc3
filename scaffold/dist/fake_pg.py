"""Recording fake collective backend (SURVEY §4 fake-backend tier).

Implements the Comm interface but records every collective (op, shape,
dtype, bytes, launch order) instead of communicating, optionally
simulating a world by averaging pre-registered peer tensors.  Verifies
bucket schedule, overlap ordering and message sizes deterministically
with ZERO GPUs and ZERO processes.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch

from .comm import Comm


@dataclass
class CollectiveRecord:
    op: str
    numel: int
    nbytes: int
    dtype: str
    order: int
    async_op: bool


class _FakeWork:
    def __init__(self, fn=None):
        self._fn = fn
        self.waited = False

    def wait(self):
        self.waited = True
        if self._fn is not None:
            self._fn()
            self._fn = None
        return True

    def is_completed(self):
        return True


class RecordingComm(Comm):
    def __init__(self, world_size: int = 1, scale_on_reduce: Optional[float] = None):
        super().__init__(group=None)
        self._world = world_size
        self._scale = scale_on_reduce
        self.records: List[CollectiveRecord] = []
        self._order = 0

    @property
    def is_distributed(self) -> bool:  # type: ignore[override]
        return True

    @property
    def world_size(self) -> int:  # type: ignore[override]
        return self._world

    @property
    def rank(self) -> int:  # type: ignore[override]
        return 0

    def _record(self, op: str, t: torch.Tensor, async_op: bool) -> CollectiveRecord:
        r = CollectiveRecord(
            op=op,
            numel=t.numel(),
            nbytes=t.numel() * t.element_size(),
            dtype=str(t.dtype),
            order=self._order,
            async_op=async_op,
        )
        self._order += 1
        self.records.append(r)
        return r

    def all_reduce(self, t: torch.Tensor, async_op: bool = False):
        self._record("all_reduce", t, async_op)

        def apply():
            if self._scale is not None:
                t.mul_(self._scale)

        w = _FakeWork(apply)
        if not async_op:
            w.wait()
        return w

    def broadcast(self, t: torch.Tensor, src: int = 0, async_op: bool = False):
        self._record("broadcast", t, async_op)
        w = _FakeWork()
        if not async_op:
            w.wait()
        return w

    def barrier(self):
        pass

    def ops(self, op: Optional[str] = None) -> List[CollectiveRecord]:
        return [r for r in self.records if op is None or r.op == op]
