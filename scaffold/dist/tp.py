"""Tensor-parallel (Megatron-style) linear primitives (SURVEY §2.4/§5.7
additive layer).

Sharding model over a TP group of `world` ranks:

  ColumnParallelLinear  W: (N, K) sharded along N -> rank r holds
      W_r: (N/world, K); y_r = x W_r^T (+ b_r).  The full activation is
      the concatenation over ranks; with `gather=False` the shard stays
      local (the usual case — the next op is row-parallel).
  RowParallelLinear     W: (N, K) sharded along K -> rank r holds
      W_r: (N, K/world); partial = x_r W_r^T, summed over ranks with a
      DIFFERENTIABLE all_reduce (backward is an identity broadcast).

A column-parallel -> row-parallel pair (the transformer MLP / attention
projection pattern) needs exactly ONE all_reduce per forward and one in
backward, with weight gradients staying rank-local.

The matmuls go through scaffold.ops.linear, so on GPU the per-shape
dispatch (hand MFMA kernel vs hipBLASLt) applies unchanged.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from .. import ops


def _initialized() -> bool:
    import torch.distributed as dist

    return dist.is_available() and dist.is_initialized()


class _CopyToTp(torch.autograd.Function):
    """Megatron's `f`: identity forward, all-reduce backward — wraps the
    REPLICATED input of a column-parallel linear so dx sums the shards'
    contributions (the loss itself is computed redundantly per rank, so
    activation grads must be reduced exactly once, here)."""

    @staticmethod
    def forward(ctx, t, group):
        ctx.group = group
        return t

    @staticmethod
    def backward(ctx, g):
        import torch.distributed as dist

        if _initialized():
            g = g.contiguous()
            dist.all_reduce(g, group=ctx.group)
        return g, None


class _ReduceFromTp(torch.autograd.Function):
    """Megatron's `g`: all-reduce forward, identity backward — sums the
    row-parallel partials; downstream grads are already replicated."""

    @staticmethod
    def forward(ctx, t, group):
        import torch.distributed as dist

        if _initialized():
            t = t.contiguous()
            dist.all_reduce(t, group=group)
        return t

    @staticmethod
    def backward(ctx, g):
        return g, None


class ColumnParallelLinear(nn.Module):
    """y_r = x @ W_r^T + b_r with W sharded along the OUTPUT dim."""

    def __init__(self, in_features: int, out_features: int, world: int,
                 rank: int, bias: bool = True, group=None):
        super().__init__()
        if out_features % world:
            raise ValueError("TP: world must divide out_features")
        self.in_features = in_features
        self.out_local = out_features // world
        self.world = world
        self.rank = rank
        self.group = group
        self.weight = nn.Parameter(torch.empty(self.out_local, in_features))
        self.bias = nn.Parameter(torch.zeros(self.out_local)) if bias else None
        nn.init.normal_(self.weight, std=0.02)

    @torch.no_grad()
    def load_full(self, w_full: torch.Tensor, b_full: Optional[torch.Tensor] = None):
        r0 = self.rank * self.out_local
        self.weight.copy_(w_full[r0:r0 + self.out_local])
        if self.bias is not None and b_full is not None:
            self.bias.copy_(b_full[r0:r0 + self.out_local])

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # backward dx all-reduce must span the TP group, not WORLD —
        # group=None here would silently sum dx across data-parallel
        # replicas in a hybrid TP x DP layout
        x = _CopyToTp.apply(x, self.group)
        return ops.linear(x, self.weight, self.bias)


class RowParallelLinear(nn.Module):
    """y = sum_r x_r @ W_r^T (+ b on rank-agnostic add) with W sharded
    along the INPUT dim; the sum is a differentiable all_reduce."""

    def __init__(self, in_features: int, out_features: int, world: int,
                 rank: int, bias: bool = True, group=None):
        super().__init__()
        if in_features % world:
            raise ValueError("TP: world must divide in_features")
        self.in_local = in_features // world
        self.world = world
        self.rank = rank
        self.group = group
        self.weight = nn.Parameter(torch.empty(out_features, self.in_local))
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None
        nn.init.normal_(self.weight, std=0.02)

    @torch.no_grad()
    def load_full(self, w_full: torch.Tensor, b_full: Optional[torch.Tensor] = None):
        c0 = self.rank * self.in_local
        self.weight.copy_(w_full[:, c0:c0 + self.in_local])
        if self.bias is not None and b_full is not None:
            self.bias.copy_(b_full)

    def forward(self, x_local: torch.Tensor) -> torch.Tensor:
        partial = ops.linear(x_local, self.weight, None)
        out = _ReduceFromTp.apply(partial, self.group)
        if self.bias is not None:
            out = out + self.bias
        return out


class TpMLP(nn.Module):
    """Column-parallel up-projection + activation + row-parallel
    down-projection: ONE all_reduce per forward (the transformer-MLP TP
    pattern).  act: "gelu" (GPT-2) or "silu-glu" would compose the same
    way with a sharded gate|up."""

    def __init__(self, d_model: int, d_ff: int, world: int, rank: int,
                 group=None):
        super().__init__()
        self.fc = ColumnParallelLinear(d_model, d_ff, world, rank, group=group)
        self.proj = RowParallelLinear(d_ff, d_model, world, rank, group=group)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        h = torch.nn.functional.gelu(self.fc(x), approximate="tanh")
        return self.proj(h)


@torch.no_grad()
def shard_llama_tp(model, rank: int, world: int, group=None):
    """Shard a built Llama IN PLACE for tensor-parallel INFERENCE:
    qkv heads and gate|up columns split across ranks (column-parallel),
    wo / w_down input-sliced (row-parallel); embeddings, norms and
    lm_head stay replicated.  The Generator completes each block with
    an all_reduce after the two row-parallel projections (it reduces
    when `attn._tp_group`-style markers are present), so per-rank decode
    touches 1/world of the attention+MLP weights — the llama-70B-class
    single-node serving layout (7 xGMI links serve the 2 all_reduces
    per block per token).

    Exactness: tests/distrib/test_tp.py proves ws=2 sharded decode ==
    the unsharded rollout token-for-token.  Training-side TP remains
    the layer-level Column/RowParallelLinear modules above.
    """
    if not hasattr(model.blocks[0].attn, "wqkv"):
        return shard_gpt2_tp(model, rank, world, group)
    for blk in model.blocks:
        if getattr(blk, "moe", None) is not None:
            raise NotImplementedError(
                "TP sharding of MoE blocks is not supported — shard the "
                "EXPERTS instead (dist.ep_moe / scaffold.dist.ep)")
        m = blk.attn
        if m.n_heads % world or m.n_kv % world:
            raise ValueError(
                f"heads {m.n_heads}/{m.n_kv} not divisible by tp={world}")
        hl = m.n_heads // world
        kl = m.n_kv // world
        d = m.head_dim
        w = m.wqkv.weight
        q = w[rank * hl * d:(rank + 1) * hl * d]
        k0 = m.n_heads * d
        k = w[k0 + rank * kl * d:k0 + (rank + 1) * kl * d]
        v0 = k0 + m.n_kv * d
        v = w[v0 + rank * kl * d:v0 + (rank + 1) * kl * d]
        m.wqkv.weight = torch.nn.Parameter(
            torch.cat([q, k, v], dim=0).contiguous())
        m.wo.weight = torch.nn.Parameter(
            m.wo.weight[:, rank * hl * d:(rank + 1) * hl * d].contiguous())
        m.n_heads = hl
        m.n_kv = kl
        m._tp_group = group
        m._tp_world = world

        gu = blk.w_gateup.weight
        F_ = gu.shape[0] // 2
        fl = F_ // world
        gate = gu[rank * fl:(rank + 1) * fl]
        up = gu[F_ + rank * fl:F_ + (rank + 1) * fl]
        blk.w_gateup.weight = torch.nn.Parameter(
            torch.cat([gate, up], dim=0).contiguous())
        blk.w_down.weight = torch.nn.Parameter(
            blk.w_down.weight[:, rank * fl:(rank + 1) * fl].contiguous())
        blk._tp_group = group
        blk._tp_world = world
    return model


@torch.no_grad()
def shard_gpt2_tp(model, rank: int, world: int, group=None):
    """GPT-2 variant of shard_llama_tp.  Differences: the qkv weight
    packs [q|k|v] each d_model rows with a BIAS (column-parallel bias
    shards with its rows), and the row-parallel projections (attn.proj,
    mlp proj) carry biases that must be added AFTER the all_reduce —
    the Generator adds them post-reduction when the markers are set
    (a partial+bias on every rank would count the bias world times)."""
    for blk in model.blocks:
        if getattr(blk, "moe", None) is not None:
            raise NotImplementedError(
                "TP sharding of MoE blocks is not supported — shard the "
                "EXPERTS instead (dist.ep_moe / scaffold.dist.ep)")
        m = blk.attn
        if m.n_heads % world:
            raise ValueError(
                f"heads {m.n_heads} not divisible by tp={world}")
        hl = m.n_heads // world
        d = m.head_dim
        dm = m.n_heads * d
        w, b = m.qkv.weight, m.qkv.bias
        rows = []
        brows = []
        for part in range(3):                 # q, k, v blocks
            lo = part * dm + rank * hl * d
            rows.append(w[lo:lo + hl * d])
            brows.append(b[lo:lo + hl * d])
        m.qkv.weight = torch.nn.Parameter(torch.cat(rows).contiguous())
        m.qkv.bias = torch.nn.Parameter(torch.cat(brows).contiguous())
        m.proj.weight = torch.nn.Parameter(
            m.proj.weight[:, rank * hl * d:(rank + 1) * hl * d].contiguous())
        # proj.bias stays replicated; added post-reduce by the Generator
        m.n_heads = hl
        m._tp_group = group
        m._tp_world = world

        F_ = blk.fc.weight.shape[0]
        fl = F_ // world
        blk.fc.weight = torch.nn.Parameter(
            blk.fc.weight[rank * fl:(rank + 1) * fl].contiguous())
        blk.fc.bias = torch.nn.Parameter(
            blk.fc.bias[rank * fl:(rank + 1) * fl].contiguous())
        blk.proj.weight = torch.nn.Parameter(
            blk.proj.weight[:, rank * fl:(rank + 1) * fl].contiguous())
        blk._tp_group = group
        blk._tp_world = world
    return model
