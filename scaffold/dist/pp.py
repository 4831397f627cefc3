"""Pipeline parallelism, GPipe-style (SURVEY §2.4 additive layer).

A model is split into `world` stages, one per rank; a global batch is
split into micro-batches.  Schedule: all micro-forwards flow down the
pipeline (activations sent rank r -> r+1), the last stage computes the
loss per micro-batch, then all micro-backwards flow back (activation
gradients sent r+1 -> r).  Weight gradients ACCUMULATE across
micro-batches on each stage, so the result is bit-identical to the
unpipelined model on the full batch (mean-loss semantics: the caller's
loss_fn should average over the micro-batch; this runner averages the
micro-losses).

Transport is explicit dist.send/recv with a manual autograd bridge at
each stage boundary (recv'd activation -> leaf with requires_grad; its
.grad is what gets sent back), which keeps the schedule visible and
works on gloo (CPU tests) and RCCL alike.
"""

from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist


def _send(t: torch.Tensor, dst: int, group=None) -> None:
    dist.send(t.contiguous(), dst=dst, group=group)


def _recv(shape, dtype, src: int, device, group=None) -> torch.Tensor:
    t = torch.empty(shape, dtype=dtype, device=device)
    dist.recv(t, src=src, group=group)
    return t


class PipelineStage:
    """One rank's slice of a GPipe pipeline.

    stage_fn: the stage's nn.Module (activations in -> activations out);
    the LAST stage's module maps activations -> per-microbatch scalar
    loss (the labels are provided to run_batch).
    """

    def __init__(self, module: torch.nn.Module, rank: int, world: int,
                 act_shape, act_dtype=torch.float32, device="cpu", group=None):
        self.module = module
        self.rank = rank
        self.world = world
        self.act_shape = tuple(act_shape)  # per-microbatch activation shape
        self.act_dtype = act_dtype
        self.device = torch.device(device)
        self.group = group

    @property
    def is_first(self) -> bool:
        return self.rank == 0

    @property
    def is_last(self) -> bool:
        return self.rank == self.world - 1

    def run_batch(self, micro_inputs: List[torch.Tensor] = None,
                  micro_labels: List[torch.Tensor] = None,
                  n_micro: int = None) -> float:
        """One pipelined forward+backward over n_micro micro-batches.
        First stage consumes micro_inputs; last stage consumes
        micro_labels and returns the mean micro-loss (0.0 elsewhere).
        Weight grads are left ACCUMULATED on each stage's module."""
        if n_micro is None:
            n_micro = len(micro_inputs) if micro_inputs else len(micro_labels)

        saved_in: List[torch.Tensor] = []   # boundary leaves (for bwd)
        saved_out: List[torch.Tensor] = []  # boundary outputs (for bwd)
        losses: List[torch.Tensor] = []

        # ---- forward wave
        for m in range(n_micro):
            if self.is_first:
                x = micro_inputs[m]
            else:
                x = _recv(self.act_shape, self.act_dtype, self.rank - 1,
                          self.device, self.group)
                x = x.requires_grad_(True)
            saved_in.append(x)
            if self.is_last:
                loss = self.module(x, micro_labels[m])
                losses.append(loss)
            else:
                y = self.module(x)
                saved_out.append(y)
                _send(y.detach(), self.rank + 1, self.group)

        # ---- backward wave (reverse order, matching the fwd stream)
        for m in reversed(range(n_micro)):
            if self.is_last:
                # mean over micro-batches
                (losses[m] / n_micro).backward()
            else:
                dy = _recv(self.act_shape, self.act_dtype, self.rank + 1,
                           self.device, self.group)
                saved_out[m].backward(dy)
            if not self.is_first:
                _send(saved_in[m].grad, self.rank - 1, self.group)

        if self.is_last:
            return float(sum(l.detach() for l in losses) / n_micro)
        return 0.0

    def run_batch_1f1b(self, micro_inputs: List[torch.Tensor] = None,
                       micro_labels: List[torch.Tensor] = None,
                       n_micro: int = None) -> float:
        """One-forward-one-backward schedule: after a rank-dependent
        warmup (world-1-rank forwards), every forward is immediately
        paired with the backward of the OLDEST outstanding micro-batch,
        so at most world-rank activations are ever live per stage —
        GPipe's run_batch keeps all n_micro.  Same math (micro-grads
        accumulate; fp32 summation ORDER differs from GPipe's reversed
        wave, so equality with the unpipelined model is allclose, not
        bitwise).  `peak_live` records the measured activation high-water
        mark for the memory claim."""
        if n_micro is None:
            n_micro = len(micro_inputs) if micro_inputs else len(micro_labels)
        pending_in: List[torch.Tensor] = []   # FIFO of boundary leaves
        pending_out: List[torch.Tensor] = []  # FIFO of boundary outputs
        loss_sum = 0.0
        self.peak_live = 0
        fwd_done = 0
        bwd_done = 0
        # 1F1B interleaves sends in BOTH directions: a blocking
        # (rendezvous) send deadlocks head-to-head with the neighbour's
        # grad send (measured: rank r in send-F(m+1) vs rank r+1 in
        # send-B(m)).  isend + deferred wait keeps the schedule's
        # pairwise channel order without the rendezvous coupling; the
        # sent tensors (detached y, x.grad) are never mutated afterwards.
        sends = []

        def _isend(t, dst):
            sends.append(dist.isend(t.contiguous(), dst=dst,
                                    group=self.group))

        def one_fwd():
            nonlocal fwd_done
            m = fwd_done
            if self.is_first:
                x = micro_inputs[m]
            else:
                x = _recv(self.act_shape, self.act_dtype, self.rank - 1,
                          self.device, self.group)
                x = x.requires_grad_(True)
            pending_in.append(x)
            if self.is_last:
                pending_out.append(self.module(x, micro_labels[m]))
            else:
                y = self.module(x)
                pending_out.append(y)
                _isend(y.detach(), self.rank + 1)
            fwd_done += 1
            self.peak_live = max(self.peak_live, len(pending_in))

        def one_bwd():
            nonlocal bwd_done, loss_sum
            x = pending_in.pop(0)
            out = pending_out.pop(0)
            if self.is_last:
                loss_sum += float(out.detach())
                (out / n_micro).backward()
            else:
                dy = _recv(self.act_shape, self.act_dtype, self.rank + 1,
                           self.device, self.group)
                out.backward(dy)
            if not self.is_first:
                _isend(x.grad, self.rank - 1)
            bwd_done += 1

        warmup = min(self.world - 1 - self.rank, n_micro)
        for _ in range(warmup):
            one_fwd()
        while fwd_done < n_micro:
            one_fwd()
            one_bwd()
        while bwd_done < n_micro:
            one_bwd()
        for w in sends:
            w.wait()
        return loss_sum / n_micro if self.is_last else 0.0
