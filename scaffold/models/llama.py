"""Llama-3 family: RMSNorm, RoPE, GQA attention, SwiGLU (contract config 5).

The llama3_8b preset (32 layers, d=4096, 32/8 heads, d_ff=14336,
V=128256) sizes the 288 GB-per-GPU HBM check: 8.0B params -> bf16
working 16 GB + fp32 master/m/v 96 GB + activations, comfortably
single-GPU DDP on MI355X.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x):
        return ops.rms_norm(x, self.weight, self.eps)


class LlamaAttention(nn.Module):
    """GQA attention with ONE fused QKV projection (q|k|v heads packed
    along the output dim -> a single large GEMM instead of three)."""

    def __init__(self, d_model: int, n_heads: int, n_kv_heads: int):
        super().__init__()
        if d_model % n_heads or n_heads % n_kv_heads:
            raise ValueError(
                f"d_model {d_model} / n_heads {n_heads} / n_kv "
                f"{n_kv_heads}: need n_heads | d_model and n_kv | n_heads")
        self.n_heads = n_heads
        self.n_kv = n_kv_heads
        self.head_dim = d_model // n_heads
        self.wqkv = nn.Linear(
            d_model, (n_heads + 2 * n_kv_heads) * self.head_dim, bias=False)
        self.wo = nn.Linear(n_heads * self.head_dim, d_model, bias=False)

    def forward(self, x, cos, sin, residual):
        B, S, _ = x.shape
        qkv = ops.linear(x, self.wqkv.weight)
        hv = qkv.view(B, S, self.n_heads + 2 * self.n_kv, self.head_dim)
        q = hv[:, :, :self.n_heads].transpose(1, 2)          # (B,H,S,D) strided
        k = hv[:, :, self.n_heads:self.n_heads + self.n_kv].transpose(1, 2)
        v = hv[:, :, self.n_heads + self.n_kv:].transpose(1, 2)
        q = ops.rope_apply(q, cos, sin)
        k = ops.rope_apply(k, cos, sin)
        o = ops.attention(q, k, v, causal=True)
        return ops.linear_residual(
            o.transpose(1, 2).reshape(B, S, -1), self.wo.weight, None, residual)


class LlamaBlock(nn.Module):
    def __init__(self, d_model: int, n_heads: int, n_kv_heads: int,
                 d_ff: int, eps: float, n_experts: int = 0, ep=None,
                 init_seed: int = 0, moe_top_k: int = 1):
        super().__init__()
        self.attn_norm = RMSNorm(d_model, eps)
        self.attn = LlamaAttention(d_model, n_heads, n_kv_heads)
        self.mlp_norm = RMSNorm(d_model, eps)
        if n_experts > 0:
            # Mixtral-class routed FFN: SwiGLU experts (packed gate|up,
            # the same layout as the dense w_gateup), sharded across
            # ranks when an ep context is present (dist.ep_moe)
            from ..dist.ep import ExpertParallelMoE
            world, rank, group = (ep if ep is not None else (1, 0, None))
            self.moe = ExpertParallelMoE(
                d_model, d_ff, n_experts, world=world, rank=rank,
                group=group,
                dispatch="alltoall" if world > 1 else "dense",
                init_seed=init_seed, top_k=moe_top_k, act="swiglu")
            if world > 1:
                for plist in (self.moe.w_up, self.moe.w_down):
                    for p_ in plist:
                        p_._no_ddp = True
        else:
            self.moe = None
            # gate and up fused into one [2*d_ff, d] GEMM; the packed
            # SwiGLU kernel consumes the gate|up halves in place
            self.w_gateup = nn.Linear(d_model, 2 * d_ff, bias=False)
            self.w_down = nn.Linear(d_ff, d_model, bias=False)

    def forward(self, x, cos, sin):
        x = self.attn(self.attn_norm(x), cos, sin, residual=x)
        h = self.mlp_norm(x)
        if self.moe is not None:
            B, S, C = h.shape
            return x + self.moe(h.reshape(B * S, C)).view(B, S, C)
        g = ops.swiglu_packed(ops.linear(h, self.w_gateup.weight))
        x = ops.linear_residual(g, self.w_down.weight, None, x)
        return x


class Llama(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        m = cfg.model
        self.act_ckpt = getattr(m, "act_ckpt", False)
        d_ff = m.d_ff or int(8 * m.d_model / 3 + 255) // 256 * 256
        n_kv = m.n_kv_heads or m.n_heads
        self.cfg = cfg
        self.embed = nn.Embedding(m.vocab_size, m.d_model)
        ep = getattr(cfg, "_ep_ctx", None)
        self.blocks = nn.ModuleList(
            LlamaBlock(m.d_model, m.n_heads, n_kv, d_ff, m.norm_eps,
                       getattr(m, "n_experts", 0), ep,
                       init_seed=getattr(cfg, "seed", 0) * 4099 + li,
                       moe_top_k=getattr(m, "moe_top_k", 1))
            for li in range(m.n_layers)
        )
        self.norm_f = RMSNorm(m.d_model, m.norm_eps)
        self.lm_head = nn.Linear(m.d_model, m.vocab_size, bias=False)
        head_dim = m.d_model // m.n_heads
        cos, sin = ops.rope_tables(
            m.max_seq_len, head_dim, m.rope_theta, "cpu",
            scaling=getattr(m, "rope_scaling", "none"),
            factor=getattr(m, "rope_factor", 8.0),
            low_freq_factor=getattr(m, "rope_low_freq_factor", 1.0),
            high_freq_factor=getattr(m, "rope_high_freq_factor", 4.0),
            orig_max_pos=getattr(m, "rope_orig_max_pos", 8192))
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        for n_, p in self.named_parameters():
            if p.dim() > 1 and ".moe." not in n_:
                nn.init.normal_(p, std=0.02)
        if ep is not None and ep[0] > 1:
            from ..dist.ep import mark_ep_checkpoint_identities

            mark_ep_checkpoint_identities(self, ep[0], ep[1])

    def forward(self, tokens: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
        B, S = tokens.shape
        cos = self.rope_cos[:S]
        sin = self.rope_sin[:S]
        x = self.embed(tokens)
        for blk in self.blocks:
            if self.act_ckpt and self.training:
                x = torch.utils.checkpoint.checkpoint(
                    blk, x, cos, sin, use_reentrant=False)
            else:
                x = blk(x, cos, sin)
        # fused chunked lm_head+CE (ops.lm_head_ce)
        return ops.lm_head_ce(self.norm_f(x), self.lm_head.weight, targets)
