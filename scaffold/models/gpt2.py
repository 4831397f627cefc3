"""GPT-2-small (124M): LayerNorm, causal attention, GELU MLP, tied embeddings.

Contract config 4.  Every hot op goes through scaffold.ops: LayerNorm
(K2), flash attention (K4), softmax-cross-entropy (K3); GEMMs through
torch.matmul (hipBLASLt) with the hand MFMA path arriving per SURVEY
§2.3 K5.
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn

from .. import ops


class LayerNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))
        self.eps = eps

    def forward(self, x):
        return ops.layer_norm(x, self.weight, self.bias, self.eps)


class CausalSelfAttention(nn.Module):
    def __init__(self, d_model: int, n_heads: int):
        super().__init__()
        if d_model % n_heads:
            raise ValueError(f"d_model {d_model} not divisible by "
                             f"n_heads {n_heads}")
        self.n_heads = n_heads
        self.head_dim = d_model // n_heads
        self.qkv = nn.Linear(d_model, 3 * d_model)
        self.proj = nn.Linear(d_model, d_model)

    def forward(self, x, residual):
        B, S, C = x.shape
        qkv = ops.linear(x, self.qkv.weight, self.qkv.bias)
        qkv = qkv.view(B, S, 3, self.n_heads, self.head_dim)
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))  # (B,H,S,D)
        o = ops.attention(q, k, v, causal=True)
        # output projection with the residual add fused into the epilogue
        return ops.linear_residual(
            o.transpose(1, 2).reshape(B, S, C),
            self.proj.weight, self.proj.bias, residual)


class Block(nn.Module):
    def __init__(self, d_model: int, n_heads: int, d_ff: int, eps: float,
                 n_experts: int = 0, ep=None, init_seed: int = 0,
                 moe_top_k: int = 1):
        super().__init__()
        self.ln1 = LayerNorm(d_model, eps)
        self.attn = CausalSelfAttention(d_model, n_heads)
        self.ln2 = LayerNorm(d_model, eps)
        if n_experts > 0:
            # top-1-routed MoE FFN (model.n_experts).  Without an ep
            # context the experts are local; with one (dist.ep_moe)
            # they shard across the group and the routed tokens travel
            # by all_to_all — expert params are marked _no_ddp so the
            # bucket engine leaves them rank-local.
            from ..dist.ep import ExpertParallelMoE
            world, rank, group = (ep if ep is not None else (1, 0, None))
            self.moe = ExpertParallelMoE(
                d_model, d_ff, n_experts, world=world, rank=rank,
                group=group,
                dispatch="alltoall" if world > 1 else "dense",
                init_seed=init_seed, top_k=moe_top_k)
            if world > 1:
                for plist in (self.moe.w_up, self.moe.w_down):
                    for p_ in plist:
                        p_._no_ddp = True
        else:
            self.moe = None
            self.fc = nn.Linear(d_model, d_ff)
            self.proj = nn.Linear(d_ff, d_model)

    def forward(self, x):
        # hot ops routed through scaffold.ops: K5 hand MFMA linears with
        # fused bias+GELU (fc) and fused residual-add (projections)
        x = self.attn(self.ln1(x), residual=x)
        h = self.ln2(x)
        if self.moe is not None:
            B, S, C = h.shape
            return x + self.moe(h.reshape(B * S, C)).view(B, S, C)
        h = ops.linear_gelu(h, self.fc.weight, self.fc.bias)
        x = ops.linear_residual(h, self.proj.weight, self.proj.bias, x)
        return x


class GPT2(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        m = cfg.model
        d_ff = m.d_ff or 4 * m.d_model
        self.act_ckpt = getattr(m, "act_ckpt", False)
        self.wte = nn.Embedding(m.vocab_size, m.d_model)
        self.wpe = nn.Embedding(m.max_seq_len, m.d_model)
        ep = getattr(cfg, "_ep_ctx", None)
        self.blocks = nn.ModuleList(
            Block(m.d_model, m.n_heads, d_ff, m.norm_eps,
                  getattr(m, "n_experts", 0), ep,
                  init_seed=getattr(cfg, "seed", 0) * 4099 + li,
                  moe_top_k=getattr(m, "moe_top_k", 1))
            for li in range(m.n_layers)
        )
        if ep is not None and ep[0] > 1:  # (world, rank, group)
            from ..dist.ep import mark_ep_checkpoint_identities

            mark_ep_checkpoint_identities(self, ep[0], ep[1])
        self.ln_f = LayerNorm(m.d_model, m.norm_eps)
        self.lm_head = nn.Linear(m.d_model, m.vocab_size, bias=False)
        if m.tie_embeddings:
            self.lm_head.weight = self.wte.weight
        self.apply(self._init)
        # scaled init for residual projections (GPT-2 recipe)
        for name, p in self.named_parameters():
            if name.endswith("proj.weight"):
                nn.init.normal_(p, std=0.02 / math.sqrt(2 * m.n_layers))

    @staticmethod
    def _init(module):
        if isinstance(module, nn.Linear):
            nn.init.normal_(module.weight, std=0.02)
            if module.bias is not None:
                nn.init.zeros_(module.bias)
        elif isinstance(module, nn.Embedding):
            nn.init.normal_(module.weight, std=0.02)

    def forward(self, tokens: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
        B, S = tokens.shape
        pos = torch.arange(S, device=tokens.device)
        x = self.wte(tokens) + self.wpe(pos)
        for blk in self.blocks:
            if self.act_ckpt and self.training:
                # no dropout anywhere in the family -> recompute is
                # bitwise-identical (tests/unit/test_act_ckpt.py)
                x = torch.utils.checkpoint.checkpoint(
                    blk, x, use_reentrant=False)
            else:
                x = blk(x)
        # fused chunked lm_head+CE: the [T, V] logits never round-trip HBM
        return ops.lm_head_ce(self.ln_f(x), self.lm_head.weight, targets)
