"""AdamW with fp32 master weights and a single fused multi-tensor step.

The model holds bf16 (or fp32) working parameters; the optimizer owns
fp32 masters + moments and updates everything in ONE multi-tensor HIP
kernel launch per step on GPU (K1, pure-bandwidth — SURVEY §2.3), which
also refreshes the bf16 working copy in the same pass (K8 folded in).
"""

from __future__ import annotations

from typing import Iterable, List

import torch

from . import ops


class MasterAdamW:
    def __init__(
        self,
        params: Iterable[torch.nn.Parameter],
        lr: float = 1e-3,
        betas=(0.9, 0.95),
        eps: float = 1e-8,
        weight_decay: float = 0.01,
        fused: bool = True,
        param_names=None,
        shard_owners=None,
    ):
        # param_names: checkpoint identities aligned with `params` (set
        # when the optimizer covers a SUBSET of the model — ZeRO-1);
        # shard_owners: the full trainable-name -> owner-rank map (same
        # deterministic partition on every rank), carried so the
        # checkpoint layer can build a consistent manifest.
        self.param_names = list(param_names) if param_names is not None else None
        self.shard_owners = dict(shard_owners) if shard_owners is not None else None
        self.params: List[torch.nn.Parameter] = [p for p in params if p.requires_grad]
        if not self.params:
            raise ValueError("no trainable parameters")
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.fused = fused
        self.step_count = 0
        self._cached = None  # prebuilt (meta_dev, n, total, bf16, out) per group
        self._cached_grad_ids = None  # grad data_ptrs the cache was built for
        self.step_dev = None  # optional device step (hipGraph replay path)
        # fp32 masters + moments.  fp32 working params ALIAS their master
        # (no copy-back pass); low-precision params get a separate master.
        self.master: List[torch.Tensor] = [
            p.data if p.dtype == torch.float32 else p.detach().to(torch.float32).clone()
            for p in self.params
        ]
        self.exp_avg = [torch.zeros_like(m) for m in self.master]
        self.exp_avg_sq = [torch.zeros_like(m) for m in self.master]

    def zero_grad(self, set_to_none: bool = False) -> None:
        for p in self.params:
            if p.grad is not None:
                if set_to_none:
                    p.grad = None
                else:
                    p.grad.zero_()

    def _build_groups(self):
        # one fused launch per homogeneous group: {bf16 working + bf16-out}
        # and {fp32 working aliasing master, no out}
        groups = {}  # (grad_dtype, needs_out) -> lists
        for i, p in enumerate(self.params):
            if p.grad is None:
                continue
            needs_out = p.dtype != torch.float32
            key = (p.grad.dtype, needs_out)
            g = groups.setdefault(key, ([], [], [], [], []))
            g[0].append(self.master[i])
            g[1].append(p.grad)
            g[2].append(self.exp_avg[i])
            g[3].append(self.exp_avg_sq[i])
            if needs_out:
                g[4].append(p.data)
        return groups

    @torch.no_grad()
    def step(self, grad_scale: float = 1.0) -> None:
        self.step_count += 1
        on_gpu = self.params[0].is_cuda and ops.has_ext()
        if on_gpu:
            # the device meta table caches raw grad POINTERS; rebuild if
            # any grad tensor identity changed since it was built (e.g.
            # zero_grad(set_to_none=True) re-materialized grads, or a
            # param had no grad at the first step) — stale pointers
            # would silently update from freed memory
            grad_ids = tuple(
                p.grad.data_ptr() if p.grad is not None else 0
                for p in self.params
            )
            if self._cached is not None and grad_ids != self._cached_grad_ids:
                self._cached = None
            if self._cached is None:
                # tensor pointers are stable across steps (grads are
                # DDP bucket views; masters/moments never reallocate):
                # build the device meta table ONCE — the per-step call
                # then carries no H2D copy and is hipGraph-capturable
                self._cached_grad_ids = grad_ids
                m = ops.require_ext()
                cached = []
                for (gdt, needs_out), (ms, gs, eas, eass, outs) in self._build_groups().items():
                    meta = m.adamw_build_meta(ms, gs, eas, eass, outs if needs_out else [])
                    total = sum((t.numel() + 3) // 4 for t in ms)
                    cached.append((meta, len(ms), total,
                                   gdt == torch.bfloat16, needs_out))
                self._cached = cached
            m = ops.require_ext()
            for meta, n, total, gbf16, has_out in self._cached:
                m.fused_adamw_run(
                    meta, n, total, self.lr, self.beta1, self.beta2, self.eps,
                    self.weight_decay, self.step_count, grad_scale,
                    gbf16, has_out, self.step_dev,
                )
            return
        for (gdt, needs_out), (masters, grads, eas, eass, outs) in self._build_groups().items():
            ops.fused_adamw_step(
                masters, grads, eas, eass, outs if needs_out else None,
                lr=self.lr, beta1=self.beta1, beta2=self.beta2, eps=self.eps,
                weight_decay=self.weight_decay, step=self.step_count,
                grad_scale=grad_scale,
            )

    def state_dict(self) -> dict:
        return {
            "step": self.step_count,
            "lr": self.lr,
            "master": self.master,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
        }

    def load_state_dict(self, sd: dict) -> None:
        self.step_count = sd["step"]
        self.lr = sd.get("lr", self.lr)
        for dst, src in zip(self.master, sd["master"]):
            dst.copy_(src.to(dst.device))
        for dst, src in zip(self.exp_avg, sd["exp_avg"]):
            dst.copy_(src.to(dst.device))
        for dst, src in zip(self.exp_avg_sq, sd["exp_avg_sq"]):
            dst.copy_(src.to(dst.device))
        # refresh working params from masters
        with torch.no_grad():
            for p, m in zip(self.params, self.master):
                p.data.copy_(m.to(p.dtype))


def build_optimizer(cfg, model: torch.nn.Module, ctx=None) -> MasterAdamW:
    o = cfg.optim
    zero1 = (getattr(cfg.dist, "zero1", False) and ctx is not None
             and ctx.initialized and ctx.world_size > 1)
    if not zero1:
        # explicit trainable identities: with frozen params (LoRA) the
        # optimizer's positional lists no longer align with
        # model.named_parameters(), so the checkpoint layer needs names
        trainable = [(getattr(p, "_ckpt_name", n), p)
                     for n, p in model.named_parameters() if p.requires_grad]
        return MasterAdamW(
            [p for _, p in trainable],
            lr=o.lr,
            betas=tuple(o.betas),
            eps=o.eps,
            weight_decay=o.weight_decay,
            fused=o.fused,
            param_names=[n for n, _ in trainable],
        )
    # ZeRO-1: deterministic size-balanced ownership over the trainable
    # params (same map on every rank); this rank's optimizer holds
    # masters/moments ONLY for its shard
    from .ckpt import layout

    trainable = [(n, p) for n, p in model.named_parameters() if p.requires_grad]
    owners = layout.partition_by_size(
        [(n, p.numel()) for n, p in trainable], ctx.world_size)
    own = [(n, p) for n, p in trainable if owners[n] == ctx.rank]
    return MasterAdamW(
        [p for _, p in own],
        lr=o.lr,
        betas=tuple(o.betas),
        eps=o.eps,
        weight_decay=o.weight_decay,
        fused=o.fused,
        param_names=[n for n, _ in own],
        shard_owners=owners,
    )
