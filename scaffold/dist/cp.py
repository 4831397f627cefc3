"""Context-parallel attention over a sequence-sharded batch
(the attention op is a single boundary whose (o, lse)
online-softmax state is exactly what a context-parallel wrapper needs —
this module composes AROUND ops.attention_lse without touching the HIP
kernel).

Sharding model: the sequence is split evenly over the CP group's ranks;
rank r holds q_r, k_r, v_r = shard r of (B, H/ Hkv, S_total, D).  Each
rank computes its queries' attention against every kv shard it may
attend (j < r: full; j == r: causal; j > r: skipped under causality)
and merges the partials EXACTLY through their logsumexp:

    lse = logsumexp_j(lse_j)        O = sum_j exp(lse_j - lse) * O_j

The kv shards are exchanged with the DIFFERENTIABLE all_gather from
torch.distributed.nn.functional, so backward reduce-scatters dk/dv to
their owner ranks automatically; d(lse_j) flows into each partial's
backward through attention_lse (see the dlse plumbing note in
scaffold/ops).  This is the all-gather form of ring attention (the
ring-pass form changes only the schedule of the same partial merges).
"""

from __future__ import annotations

from typing import List

import torch

from .. import ops


def _merge(partials: List[torch.Tensor], lses: List[torch.Tensor]) -> torch.Tensor:
    """Exact lse-weighted merge of per-shard attention partials."""
    if len(partials) == 1:
        return partials[0]
    lse_all = torch.stack(lses, dim=0)                     # (J, B, H, S)
    lse_tot = torch.logsumexp(lse_all, dim=0)              # (B, H, S)
    w = torch.exp(lse_all - lse_tot.unsqueeze(0))          # (J, B, H, S)
    out = torch.zeros_like(partials[0], dtype=torch.float32)
    for j, oj in enumerate(partials):
        out = out + w[j].unsqueeze(-1) * oj.float()
    return out.to(partials[0].dtype)


def cp_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 rank: int, world: int, causal: bool = True,
                 group=None) -> torch.Tensor:
    """Attention for THIS rank's query shard against the group's full
    (sharded) kv sequence.  q/k/v: (B, H|Hkv, S_local, D)."""
    if world <= 1:
        return ops.attention(q, k, v, causal)

    from torch.distributed import nn as dist_nn

    k_shards = list(dist_nn.functional.all_gather(k, group=group))
    v_shards = list(dist_nn.functional.all_gather(v, group=group))

    partials, lses = [], []
    last = rank if causal else world - 1
    for j in range(0, last + 1):
        o_j, lse_j = ops.attention_lse(
            q, k_shards[j], v_shards[j],
            causal=(causal and j == rank))
        partials.append(o_j)
        lses.append(lse_j)
    return _merge(partials, lses)


class _RingShift(torch.autograd.Function):
    """Differentiable ring shift: out_r = in_{(r-1) mod world}.

    Forward sends this rank's tensor to (r+1) and receives from (r-1);
    backward reverses the ring (grad of the received tensor returns to
    its sender).  Point-to-point send/recv — exactly the xGMI transport
    shape (7 p2p links per GPU); at any moment only ONE extra kv shard
    is live per rank instead of the all-gather form's world shards, and
    the next shard's transfer overlaps this shard's attention partial.
    """

    @staticmethod
    def forward(ctx, t, rank, world, group):
        import torch.distributed as dist

        ctx.ring = (rank, world, group)
        nxt = (rank + 1) % world
        prv = (rank - 1 + world) % world
        out = torch.empty_like(t)
        send = t.contiguous()
        req = dist.isend(send, dst=nxt, group=group)
        dist.recv(out, src=prv, group=group)
        req.wait()
        return out

    @staticmethod
    def backward(ctx, g):
        import torch.distributed as dist

        rank, world, group = ctx.ring
        nxt = (rank + 1) % world
        prv = (rank - 1 + world) % world
        gin = torch.empty_like(g)
        req = dist.isend(g.contiguous(), dst=prv, group=group)
        dist.recv(gin, src=nxt, group=group)
        req.wait()
        return gin, None, None, None


def cp_attention_ring(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                      rank: int, world: int, causal: bool = True,
                      group=None) -> torch.Tensor:
    """Ring-pass schedule of the same lse-merge as cp_attention: kv
    shards travel around the ring one hop per step while each rank
    accumulates its partials — peak extra memory is ONE kv shard (vs
    world shards for the all-gather form) and each hop's transfer
    overlaps the previous shard's attention compute.  Math identical
    (exact lse merge); dk/dv return to their owners through the
    reversed ring in backward."""
    if world <= 1:
        return ops.attention(q, k, v, causal)

    kv = torch.stack([k, v], dim=0)
    partials, lses = [], []
    for s in range(world):
        j = (rank - s + world) % world
        if (not causal) or j <= rank:
            o_j, lse_j = ops.attention_lse(
                q, kv[0], kv[1], causal=(causal and j == rank))
            partials.append(o_j)
            lses.append(lse_j)
        if s < world - 1:
            kv = _RingShift.apply(kv, rank, world, group)
    out = _merge(partials, lses)
    if causal:
        # Tie the final kv into EVERY rank's graph (zero contribution):
        # a rank that skipped late shards under causality would never
        # fire the shift chain's backward, while its peers block on the
        # matching p2p sends of dk/dv — the ring must run symmetrically
        # in backward on all ranks.
        out = out + (kv.flatten()[0] * 0).to(out.dtype)
    return out


def ulysses_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                      rank: int, world: int, causal: bool = True,
                      group=None) -> torch.Tensor:
    """Ulysses-style sequence parallelism: all-to-all
    swaps the sharding axis from SEQUENCE to HEADS, each rank runs
    ordinary full-sequence attention on H/world (and Hkv/world kv)
    heads, and the inverse all-to-all restores sequence sharding.
    Exact by construction; requires world | H and world | Hkv.
    q/k/v: (B, H|Hkv, S_local, D)."""
    if world <= 1:
        return ops.attention(q, k, v, causal)
    if q.shape[1] % world or k.shape[1] % world:
        raise ValueError("ulysses: world must divide both H and Hkv")

    from torch.distributed import nn as dist_nn

    def seq_to_heads(t):
        # shard heads, gather sequence: rank r keeps head chunk r of
        # every rank's sequence shard, concatenated in rank (=sequence)
        # order
        chunks = list(t.chunk(world, dim=1))
        outs = [torch.empty_like(c) for c in chunks]
        outs = dist_nn.functional.all_to_all(outs, chunks, group=group)
        return torch.cat(list(outs), dim=2)

    def heads_to_seq(t):
        chunks = list(t.chunk(world, dim=2))
        outs = [torch.empty_like(c) for c in chunks]
        outs = dist_nn.functional.all_to_all(outs, chunks, group=group)
        return torch.cat(list(outs), dim=1)

    o = ops.attention(seq_to_heads(q), seq_to_heads(k), seq_to_heads(v), causal)
    return heads_to_seq(o)
