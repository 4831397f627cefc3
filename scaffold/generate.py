"""Autoregressive generation with per-layer KV caches (the serving face
of the scaffold).

Prefill runs the prompt through the normal block math (the flash
attention op on GPU) while recording K/V; each decode step attends the
single new query against the cache.  The decode GEMVs take the library
path — the hand decode_gemv kernel measured below hipBLASLt in-graph
and is parked behind SCAFFOLD_DECODE_GEMV=1 (NOTES r2 serving ladder).
Both model families (GPT-2 LayerNorm/GELU, Llama RMSNorm/RoPE/GQA/
SwiGLU); greedy / temperature / top-k / top-p sampling with optional
EOS; hipGraph-replayed static decode (generate_graphed); per-row
variable-position decode for continuous batching (decode_step_varpos);
speculative decoding (generate_speculative); TP-sharded models reduce
through _tp_reduce markers (dist.tp.shard_llama_tp).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from . import ops
from .models.gpt2 import GPT2
from .models.llama import Llama


def _sdpa(q, k, v, causal):
    """(B,H,Sq,D) x (B,Hkv,Skv,D) library attention with GQA expand."""
    Hq, Hkv = q.shape[1], k.shape[1]
    if Hkv != Hq:
        rep = Hq // Hkv
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    scale = 1.0 / math.sqrt(q.shape[-1])
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    if causal and q.shape[2] > 1:
        mask = torch.ones(q.shape[2], k.shape[2], dtype=torch.bool,
                          device=q.device).triu(k.shape[2] - q.shape[2] + 1)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return (p @ v.float()).to(q.dtype)


class KVCache:
    """Preallocated per-layer KV buffers.

    append() slice-assigns into the reserved region and returns views of
    the filled prefix — the r1 torch.cat form reallocated and copied the
    WHOLE cache every decode step per layer (O(S^2) traffic and
    allocator churn over a generation)."""

    def __init__(self, n_layers: int, B: int, n_kv: int, max_len: int,
                 head_dim: int, device, dtype):
        self.n_layers = n_layers
        # zeros, not empty: the padded-cache attention paths
        # (_sdpa_static / _block_varpos) compute p @ v BEFORE masking
        # limits which rows matter, and softmax-zero times an
        # uninitialized inf/NaN is NaN — freed CPU pages really do hold
        # such bit patterns (caught by the continuous-batching tests)
        self.k = [torch.zeros(B, n_kv, max_len, head_dim, device=device,
                              dtype=dtype) for _ in range(n_layers)]
        self.v = [torch.zeros_like(self.k[0]) for _ in range(n_layers)]
        self.len = 0

    def append(self, i: int, k: torch.Tensor, v: torch.Tensor
               ) -> Tuple[torch.Tensor, torch.Tensor]:
        S = k.shape[2]
        start = self.len
        self.k[i][:, :, start:start + S] = k
        self.v[i][:, :, start:start + S] = v
        if i == self.n_layers - 1:   # all layers appended for this step
            self.len = start + S
        return (self.k[i][:, :, :start + S], self.v[i][:, :, :start + S])

    @property
    def seq_len(self) -> int:
        return self.len


def _tp_reduce(t: torch.Tensor, holder) -> torch.Tensor:
    """Sum the row-parallel partial across the TP group when `holder`
    carries the shard markers (dist.tp.shard_llama_tp); no-op
    otherwise."""
    if getattr(holder, "_tp_world", 1) > 1:
        import torch.distributed as dist

        if dist.is_initialized():
            dist.all_reduce(t, group=getattr(holder, "_tp_group", None))
    return t


class Generator:
    """Greedy / temperature decoding for the scaffold's LM families."""

    def __init__(self, model: torch.nn.Module):
        self.m = model
        self.is_llama = isinstance(model, Llama)
        if not (self.is_llama or isinstance(model, GPT2)):
            raise TypeError(f"Generator: unsupported model {type(model).__name__}")

    # ---- per-family block math with external KV ------------------------
    def _gpt2_block(self, blk, x, cache: KVCache, i: int, pos0: int):
        m = blk.attn
        B, S, C = x.shape
        h = ops.layer_norm(x, blk.ln1.weight, blk.ln1.bias, blk.ln1.eps)
        qkv = F.linear(h, m.qkv.weight, m.qkv.bias).view(
            B, S, 3, m.n_heads, m.head_dim)
        q, k, v = (qkv[:, :, j].transpose(1, 2) for j in range(3))
        k, v = cache.append(i, k, v)
        o = _sdpa(q, k, v, causal=True)
        o2 = o.transpose(1, 2).reshape(B, S, -1)
        if getattr(m, "_tp_world", 1) > 1:
            # row-parallel: reduce the partial, THEN add the replicated
            # bias (inside-linear bias would be summed world times)
            x = x + (_tp_reduce(F.linear(o2, m.proj.weight), m)
                     + m.proj.bias)
            h = ops.layer_norm(x, blk.ln2.weight, blk.ln2.bias, blk.ln2.eps)
            h = F.gelu(F.linear(h, blk.fc.weight, blk.fc.bias),
                       approximate="tanh")
            return x + (_tp_reduce(F.linear(h, blk.proj.weight), blk)
                        + blk.proj.bias)
        x = x + F.linear(o2, m.proj.weight, m.proj.bias)
        h = ops.layer_norm(x, blk.ln2.weight, blk.ln2.bias, blk.ln2.eps)
        if getattr(blk, "moe", None) is not None:
            B2, S2, C2 = h.shape
            return x + blk.moe(h.reshape(B2 * S2, C2)).view(B2, S2, C2)
        h = F.gelu(F.linear(h, blk.fc.weight, blk.fc.bias), approximate="tanh")
        return x + F.linear(h, blk.proj.weight, blk.proj.bias)

    def _llama_block(self, blk, x, cache: KVCache, i: int, pos0: int):
        m = blk.attn
        B, S, _ = x.shape
        h = ops.rms_norm(x, blk.attn_norm.weight, blk.attn_norm.eps)
        qkv = F.linear(h, m.wqkv.weight).view(
            B, S, m.n_heads + 2 * m.n_kv, m.head_dim)
        q = qkv[:, :, :m.n_heads].transpose(1, 2)
        k = qkv[:, :, m.n_heads:m.n_heads + m.n_kv].transpose(1, 2)
        v = qkv[:, :, m.n_heads + m.n_kv:].transpose(1, 2)
        cos = self.m.rope_cos[pos0:pos0 + S].float()
        sin = self.m.rope_sin[pos0:pos0 + S].float()
        q = ops._rope_eager(q, cos, sin) if not q.is_cuda else ops.rope_apply(q, cos, sin)
        k = ops._rope_eager(k, cos, sin) if not k.is_cuda else ops.rope_apply(k, cos, sin)
        k, v = cache.append(i, k, v)
        o = _sdpa(q, k, v, causal=True)
        x = x + _tp_reduce(
            F.linear(o.transpose(1, 2).reshape(B, S, -1), m.wo.weight), m)
        h = ops.rms_norm(x, blk.mlp_norm.weight, blk.mlp_norm.eps)
        if getattr(blk, "moe", None) is not None:
            B2, S2, C2 = h.shape
            return x + blk.moe(h.reshape(B2 * S2, C2)).view(B2, S2, C2)
        gu = F.linear(h, blk.w_gateup.weight)
        Fh = gu.shape[-1] // 2
        g = F.silu(gu[..., :Fh]) * gu[..., Fh:]
        return x + _tp_reduce(F.linear(g, blk.w_down.weight), blk)

    def _forward_tokens(self, tokens: torch.Tensor, cache: KVCache,
                        pos0: int, all_logits: bool = False) -> torch.Tensor:
        """tokens (B, S) at positions [pos0, pos0+S) -> logits (B, V),
        or (B, S, V) for every position when all_logits (the speculative
        verify pass needs the target's prediction AFTER each draft)."""
        m = self.m
        if self.is_llama:
            x = m.embed(tokens)
            for i, blk in enumerate(m.blocks):
                x = self._llama_block(blk, x, cache, i, pos0)
            x = ops.rms_norm(x, m.norm_f.weight, m.norm_f.eps)
        else:
            pos = torch.arange(pos0, pos0 + tokens.shape[1], device=tokens.device)
            x = m.wte(tokens) + m.wpe(pos)
            for i, blk in enumerate(m.blocks):
                x = self._gpt2_block(blk, x, cache, i, pos0)
            x = ops.layer_norm(x, m.ln_f.weight, m.ln_f.bias, m.ln_f.eps)
        if all_logits:
            return F.linear(x, m.lm_head.weight)
        return F.linear(x[:, -1], m.lm_head.weight)

    @staticmethod
    def _sample(logits: torch.Tensor, temperature: float, top_k: int,
                top_p: float,
                generator: Optional[torch.Generator]) -> torch.Tensor:
        """(B, V) logits -> (B, 1) next ids.  temperature 0 = greedy;
        top_k keeps the k best logits, top_p keeps the smallest nucleus
        with cumulative probability >= top_p (both standard serving
        truncations; composable)."""
        # tiny temperatures overflow logits/T to inf (softmax -> NaN ->
        # multinomial crash; found by the hypothesis edge-case sweep):
        # below 1e-5 the distribution is one-hot anyway, so go greedy
        if temperature <= 1e-5:
            return logits.argmax(dim=-1, keepdim=True)
        lg = logits.float() / temperature
        if top_k > 0 and top_k < lg.shape[-1]:
            kth = lg.topk(top_k, dim=-1).values[:, -1:]
            lg = lg.masked_fill(lg < kth, float("-inf"))
        if 0.0 < top_p < 1.0:
            srt, idx = lg.sort(dim=-1, descending=True)
            cum = torch.softmax(srt, dim=-1).cumsum(dim=-1)
            # keep tokens whose PRECEDING mass is < top_p; the top-1
            # token is kept UNCONDITIONALLY (a sub-float32 top_p
            # underflows the comparison to >= 0 and would drop every
            # token -> NaN probs; found by the hypothesis sweep)
            drop_sorted = (cum - torch.softmax(srt, dim=-1)) >= top_p
            drop_sorted[:, 0] = False
            drop = drop_sorted.scatter(1, idx, drop_sorted)
            lg = lg.masked_fill(drop, float("-inf"))
        probs = torch.softmax(lg, dim=-1)
        return torch.multinomial(probs, 1, generator=generator)

    @torch.no_grad()
    def generate(self, prompt: torch.Tensor, max_new_tokens: int,
                 temperature: float = 0.0, top_k: int = 0,
                 top_p: float = 0.0, eos_token_id: Optional[int] = None,
                 generator: Optional[torch.Generator] = None) -> torch.Tensor:
        """prompt (B, S0) int64 -> (B, S0 + max_new_tokens).  With
        eos_token_id, rows freeze at EOS (subsequent positions repeat
        it) and decoding stops early once every row has finished."""
        n_layers = len(self.m.blocks)
        B, S0 = prompt.shape
        attn0 = self.m.blocks[0].attn
        n_kv = attn0.n_kv if self.is_llama else attn0.n_heads
        dtype = next(self.m.parameters()).dtype
        cache = KVCache(n_layers, B, n_kv, S0 + max_new_tokens,
                        attn0.head_dim, prompt.device, dtype)
        out = torch.empty(B, S0 + max_new_tokens, dtype=prompt.dtype,
                          device=prompt.device)
        out[:, :S0] = prompt
        logits = self._forward_tokens(prompt, cache, 0)   # prefill
        done = torch.zeros(B, dtype=torch.bool, device=prompt.device)
        for t in range(max_new_tokens):
            nxt = self._sample(logits, temperature, top_k, top_p, generator)
            if eos_token_id is not None:
                nxt = torch.where(done.unsqueeze(1),
                                  torch.full_like(nxt, eos_token_id), nxt)
                done = done | (nxt[:, 0] == eos_token_id)
            out[:, S0 + t] = nxt[:, 0]
            if eos_token_id is not None and bool(done.all()):
                out[:, S0 + t + 1:] = eos_token_id
                break
            logits = self._forward_tokens(nxt, cache, cache.seq_len)
        return out


    def _lin1(self, x, weight, bias=None):
        """Single-token linear.  MEASURED: the hand decode_gemv kernel
        (2 rows/wave, 4-deep unroll) reached ~2 TB/s but hipBLASLt's
        skinny-M GEMVs are still faster inside the graphed step (llama
        graphed 437 vs 405 tok/s, gpt2 6117 vs 5374) — so the library
        path stays; the kernel is parked behind SCAFFOLD_DECODE_GEMV=1
        with its numerics test (NOTES.md r2 serving)."""
        import os

        B = x.shape[0]
        if (x.is_cuda and B <= 8 and ops.has_ext()
                and os.environ.get("SCAFFOLD_DECODE_GEMV") == "1"):
            y = ops.require_ext().decode_gemv(
                x.reshape(B, -1).contiguous(), weight, bias)
            return y.view(B, 1, -1)
        return F.linear(x, weight, bias)

    # ---- static-shape decode step (hipGraph-replayable) ---------------
    def _sdpa_static(self, q, k, v, valid_to):
        """Attention of a single query over a PADDED cache: rows at
        index >= valid_to (a device scalar tensor) are masked, so all
        shapes are replay-static."""
        Hq, Hkv = q.shape[1], k.shape[1]
        if Hkv != Hq:
            rep = Hq // Hkv
            k = k.repeat_interleave(rep, dim=1)
            v = v.repeat_interleave(rep, dim=1)
        scale = 1.0 / math.sqrt(q.shape[-1])
        s = (q.float() @ k.float().transpose(-1, -2)) * scale  # (B,H,1,L)
        idx = torch.arange(k.shape[2], device=k.device)
        s = s.masked_fill(idx >= valid_to, float("-inf"))
        p = torch.softmax(s, dim=-1)
        return (p @ v.float()).to(q.dtype)

    def _block_static(self, blk, x, cache: KVCache, i: int, pos_dev):
        """One decode block over the padded cache; pos_dev (device int64
        scalar tensor) is the write position == tokens so far."""
        if self.is_llama:
            m = blk.attn
            B = x.shape[0]
            h = ops.rms_norm(x, blk.attn_norm.weight, blk.attn_norm.eps)
            qkv = self._lin1(h, m.wqkv.weight).view(
                B, 1, m.n_heads + 2 * m.n_kv, m.head_dim)
            q = qkv[:, :, :m.n_heads].transpose(1, 2)
            k = qkv[:, :, m.n_heads:m.n_heads + m.n_kv].transpose(1, 2)
            v = qkv[:, :, m.n_heads + m.n_kv:].transpose(1, 2)
            cos = self.m.rope_cos.index_select(0, pos_dev).float()
            sin = self.m.rope_sin.index_select(0, pos_dev).float()
            q = (ops.rope_apply(q, cos, sin) if q.is_cuda
                 else ops._rope_eager(q, cos, sin))
            k = (ops.rope_apply(k, cos, sin) if k.is_cuda
                 else ops._rope_eager(k, cos, sin))
            cache.k[i].index_copy_(2, pos_dev, k)
            cache.v[i].index_copy_(2, pos_dev, v)
            o = self._sdpa_static(q, cache.k[i], cache.v[i], pos_dev + 1)
            x = x + self._lin1(o.transpose(1, 2).reshape(B, 1, -1), m.wo.weight)
            h = ops.rms_norm(x, blk.mlp_norm.weight, blk.mlp_norm.eps)
            if getattr(blk, "moe", None) is not None:
                return x + blk.moe(h.reshape(B, -1)).view(B, 1, -1)
            gu = self._lin1(h, blk.w_gateup.weight)
            Fh = gu.shape[-1] // 2
            g = F.silu(gu[..., :Fh]) * gu[..., Fh:]
            return x + self._lin1(g, blk.w_down.weight)
        m = blk.attn
        B = x.shape[0]
        h = ops.layer_norm(x, blk.ln1.weight, blk.ln1.bias, blk.ln1.eps)
        qkv = self._lin1(h, m.qkv.weight, m.qkv.bias).view(
            B, 1, 3, m.n_heads, m.head_dim)
        q, k, v = (qkv[:, :, j].transpose(1, 2) for j in range(3))
        cache.k[i].index_copy_(2, pos_dev, k)
        cache.v[i].index_copy_(2, pos_dev, v)
        o = self._sdpa_static(q, cache.k[i], cache.v[i], pos_dev + 1)
        x = x + self._lin1(o.transpose(1, 2).reshape(B, 1, -1),
                           m.proj.weight, m.proj.bias)
        h = ops.layer_norm(x, blk.ln2.weight, blk.ln2.bias, blk.ln2.eps)
        if getattr(blk, "moe", None) is not None:
            B2 = h.shape[0]
            return x + blk.moe(h.reshape(B2, -1)).view(B2, 1, -1)
        h = F.gelu(self._lin1(h, blk.fc.weight, blk.fc.bias),
                   approximate="tanh")
        return x + self._lin1(h, blk.proj.weight, blk.proj.bias)

    def _decode_step_static(self, cur_tok, cache: KVCache, pos_dev, out_dev):
        """ONE greedy decode step with fully static shapes: embeds the
        static cur_tok buffer, runs the blocks over the padded cache,
        argmaxes, writes the token at pos_dev+1 of out_dev, advances
        cur_tok and pos_dev IN PLACE (graph-replayable)."""
        m = self.m
        if self.is_llama:
            x = m.embed(cur_tok)
        else:
            x = m.wte(cur_tok) + m.wpe(pos_dev).unsqueeze(0)
        for i, blk in enumerate(m.blocks):
            x = self._block_static(blk, x, cache, i, pos_dev)
        x = (ops.rms_norm(x, m.norm_f.weight, m.norm_f.eps) if self.is_llama
             else ops.layer_norm(x, m.ln_f.weight, m.ln_f.bias, m.ln_f.eps))
        logits = self._lin1(x, m.lm_head.weight)[:, 0]
        nxt = logits.argmax(dim=-1, keepdim=True)
        out_dev.scatter_(1, (pos_dev + 1).expand(nxt.shape), nxt)
        cur_tok.copy_(nxt)
        pos_dev.add_(1)

    @torch.no_grad()
    def generate_speculative(self, prompt: torch.Tensor,
                             max_new_tokens: int, draft: "Generator",
                             k: int = 4, eos_token_id: Optional[int] = None,
                             stats: Optional[dict] = None) -> torch.Tensor:
        """Greedy speculative decoding: `draft` proposes k tokens, ONE
        target forward over the k-token window verifies them, and the
        longest matching prefix is accepted plus the target's own next
        token — output is EXACTLY Generator.generate's greedy rollout
        regardless of draft quality (only the number of target forwards
        changes).  B=1 (per-row acceptance lengths diverge in a batch).

        On MI355X the target decode step is weight-read-bound (NOTES r2
        serving), so accepting m tokens per verify cuts target weight
        reads ~m-fold; the draft adds its own (much smaller) reads.
        `stats["accepted"]/["rounds"]` expose the acceptance rate."""
        if prompt.shape[0] != 1:
            raise ValueError("speculative decode is B=1 (per-row "
                             "acceptance lengths diverge in a batch)")
        if getattr(self.m.blocks[0].attn, "_tp_world", 1) > 1:
            # a lone rank issuing the TP all_reduces would deadlock the
            # group; speculative x TP needs the command-replay protocol
            raise NotImplementedError(
                "speculative decode over a TP-sharded target is not "
                "supported")
        n_layers = len(self.m.blocks)
        attn0 = self.m.blocks[0].attn
        n_kv = attn0.n_kv if self.is_llama else attn0.n_heads
        dtype = next(self.m.parameters()).dtype
        S0 = prompt.shape[1]
        L = S0 + max_new_tokens + k + 1   # headroom for a rolled-back tail
        cache = KVCache(n_layers, 1, n_kv, L, attn0.head_dim,
                        prompt.device, dtype)
        d_attn = draft.m.blocks[0].attn
        d_kv = d_attn.n_kv if draft.is_llama else d_attn.n_heads
        d_dtype = next(draft.m.parameters()).dtype
        dcache = KVCache(len(draft.m.blocks), 1, d_kv, L, d_attn.head_dim,
                         prompt.device, d_dtype)

        out = [int(t) for t in prompt[0]]
        # prefill both; the target's prefill logits give token S0
        logits = self._forward_tokens(prompt, cache, 0)
        draft._forward_tokens(prompt, dcache, 0)
        out.append(int(logits.argmax()))
        produced = 1
        rounds = accepted_total = 0
        done = eos_token_id is not None and out[-1] == eos_token_id
        while produced < max_new_tokens and not done:
            kk = min(k, max_new_tokens - produced)
            base = len(out) - 1          # index of the newest token
            # --- draft proposes kk tokens from its own cache
            dtoks = []
            cur = torch.tensor([[out[base]]], device=prompt.device)
            # draft's cache currently holds out[:dcache.len]; catch it up
            # to the target (everything except the newest token)
            if dcache.len < base:
                catchup = torch.tensor([out[dcache.len:base]],
                                       device=prompt.device)
                draft._forward_tokens(catchup, dcache, dcache.len)
            for _ in range(kk):
                dlog = draft._forward_tokens(cur, dcache, dcache.len)
                nxt = int(dlog.argmax())
                dtoks.append(nxt)
                cur = torch.tensor([[nxt]], device=prompt.device)
            # --- one target forward over [newest, d1..d_{kk-1}]
            window = torch.tensor([[out[base]] + dtoks[:-1]],
                                  device=prompt.device)
            tlog = self._forward_tokens(window, cache, cache.len,
                                        all_logits=True)[0]
            targets = tlog.argmax(dim=-1).tolist()   # target's choice AFTER each
            m_acc = 0
            while m_acc < kk and targets[m_acc] == dtoks[m_acc]:
                m_acc += 1
            # accept d1..d_m; if a mismatch stopped us, the target's own
            # token at the mismatch position is the correction
            new = dtoks[:m_acc]
            if m_acc < kk:
                new = new + [targets[m_acc]]
            new = new[:max_new_tokens - produced]
            if eos_token_id is not None and eos_token_id in new:
                new = new[:new.index(eos_token_id) + 1]
                done = True
            out.extend(new)
            produced += len(new)
            rounds += 1
            accepted_total += m_acc
            # roll back the caches past what is actually accepted: the
            # target consumed the whole window; keep entries only for
            # tokens that are now part of `out` minus the newest
            cache.len = len(out) - 1
            dcache.len = min(dcache.len, len(out) - 1)
        if stats is not None:
            stats["rounds"] = rounds
            stats["accepted"] = accepted_total
        if eos_token_id is not None and produced < max_new_tokens:
            # match generate(): frozen rows repeat eos to full length
            out.extend([eos_token_id] * (max_new_tokens - produced))
        return torch.tensor([out[:S0 + max_new_tokens]],
                            device=prompt.device)

    # ---- variable-position decode (continuous batching) ---------------
    def _rope_rows(self, x: torch.Tensor, pos: torch.Tensor) -> torch.Tensor:
        """Per-ROW RoPE: x (B, H, 1, D) where row b sits at position
        pos[b] (unlike the shared-table paths, every slot of a
        continuous batch is at a different depth)."""
        cos = self.m.rope_cos.index_select(0, pos).float()   # (B, D/2)
        sin = self.m.rope_sin.index_select(0, pos).float()
        c = cos.view(-1, 1, 1, cos.shape[-1])
        sn = sin.view(-1, 1, 1, sin.shape[-1])
        xf = x.float()
        x1, x2 = xf[..., 0::2], xf[..., 1::2]
        out = torch.empty_like(xf)
        out[..., 0::2] = x1 * c - x2 * sn
        out[..., 1::2] = x1 * sn + x2 * c
        return out.to(x.dtype)

    def _block_varpos(self, blk, x, cache: KVCache, i: int,
                      pos: torch.Tensor):
        """One decode block where row b's token sits at position pos[b]
        ((B,) int64): per-row cache scatter, per-row causal mask, per-row
        RoPE.  The serving continuous batcher drives this; semantics per
        slot are identical to _block_static at that slot's depth
        (equivalence tested against Generator.generate)."""
        B = x.shape[0]
        if self.is_llama:
            m = blk.attn
            h = ops.rms_norm(x, blk.attn_norm.weight, blk.attn_norm.eps)
            qkv = self._lin1(h, m.wqkv.weight).view(
                B, 1, m.n_heads + 2 * m.n_kv, m.head_dim)
            q = qkv[:, :, :m.n_heads].transpose(1, 2)
            k = qkv[:, :, m.n_heads:m.n_heads + m.n_kv].transpose(1, 2)
            v = qkv[:, :, m.n_heads + m.n_kv:].transpose(1, 2)
            q = self._rope_rows(q, pos)
            k = self._rope_rows(k, pos)
            idx = pos.view(B, 1, 1, 1).expand(B, m.n_kv, 1, m.head_dim)
            cache.k[i].scatter_(2, idx, k)
            cache.v[i].scatter_(2, idx, v)
            o = self._sdpa_static(q, cache.k[i], cache.v[i],
                                  (pos + 1).view(B, 1, 1, 1))
            x = x + _tp_reduce(
                self._lin1(o.transpose(1, 2).reshape(B, 1, -1),
                           m.wo.weight), m)
            h = ops.rms_norm(x, blk.mlp_norm.weight, blk.mlp_norm.eps)
            if getattr(blk, "moe", None) is not None:
                return x + blk.moe(h.reshape(B, -1)).view(B, 1, -1)
            gu = self._lin1(h, blk.w_gateup.weight)
            Fh = gu.shape[-1] // 2
            g = F.silu(gu[..., :Fh]) * gu[..., Fh:]
            return x + _tp_reduce(self._lin1(g, blk.w_down.weight), blk)
        m = blk.attn
        h = ops.layer_norm(x, blk.ln1.weight, blk.ln1.bias, blk.ln1.eps)
        qkv = self._lin1(h, m.qkv.weight, m.qkv.bias).view(
            B, 1, 3, m.n_heads, m.head_dim)
        q, k, v = (qkv[:, :, j].transpose(1, 2) for j in range(3))
        idx = pos.view(B, 1, 1, 1).expand(B, m.n_heads, 1, m.head_dim)
        cache.k[i].scatter_(2, idx, k)
        cache.v[i].scatter_(2, idx, v)
        o = self._sdpa_static(q, cache.k[i], cache.v[i],
                              (pos + 1).view(B, 1, 1, 1))
        o2 = o.transpose(1, 2).reshape(B, 1, -1)
        if getattr(m, "_tp_world", 1) > 1:
            x = x + (_tp_reduce(self._lin1(o2, m.proj.weight), m)
                     + m.proj.bias)
            h = ops.layer_norm(x, blk.ln2.weight, blk.ln2.bias, blk.ln2.eps)
            h = F.gelu(self._lin1(h, blk.fc.weight, blk.fc.bias),
                       approximate="tanh")
            return x + (_tp_reduce(self._lin1(h, blk.proj.weight), blk)
                        + blk.proj.bias)
        x = x + self._lin1(o2, m.proj.weight, m.proj.bias)
        h = ops.layer_norm(x, blk.ln2.weight, blk.ln2.bias, blk.ln2.eps)
        if getattr(blk, "moe", None) is not None:
            B2 = h.shape[0]
            return x + blk.moe(h.reshape(B2, -1)).view(B2, 1, -1)
        h = F.gelu(self._lin1(h, blk.fc.weight, blk.fc.bias),
                   approximate="tanh")
        return x + self._lin1(h, blk.proj.weight, blk.proj.bias)

    @torch.no_grad()
    def decode_step_varpos(self, cur_tok: torch.Tensor, cache: KVCache,
                           pos: torch.Tensor) -> torch.Tensor:
        """cur_tok (B, 1) int64, pos (B,) int64 (per-row position of
        cur_tok) -> next-token logits (B, V).  Writes cur_tok's kv at
        pos[b]; caller owns sampling and position advance."""
        m = self.m
        if self.is_llama:
            x = m.embed(cur_tok)
        else:
            x = m.wte(cur_tok) + m.wpe(pos).unsqueeze(1)
        for i, blk in enumerate(m.blocks):
            x = self._block_varpos(blk, x, cache, i, pos)
        x = (ops.rms_norm(x, m.norm_f.weight, m.norm_f.eps) if self.is_llama
             else ops.layer_norm(x, m.ln_f.weight, m.ln_f.bias, m.ln_f.eps))
        return self._lin1(x, m.lm_head.weight)[:, 0]

    @torch.no_grad()
    def generate_graphed(self, prompt: torch.Tensor, max_new_tokens: int,
                         use_graph: bool = True) -> torch.Tensor:
        """Greedy generation with a hipGraph-replayed decode step: after
        the (dynamic-shape) prefill, every per-token forward is ONE
        graph replay over static buffers — cache writes, attention
        masking, argmax and position advance all run device-side
        against a device position scalar.  use_graph=False runs the
        same static-shape step eagerly (the CPU-testable path)."""
        n_layers = len(self.m.blocks)
        B, S0 = prompt.shape
        attn0 = self.m.blocks[0].attn
        if getattr(attn0, "_tp_world", 1) > 1:
            # the static step has no TP reductions — failing loudly
            # beats silently wrong tokens; TP decode runs through
            # generate() / ContinuousBatcher (both reduce)
            raise NotImplementedError(
                "generate_graphed does not support TP-sharded models")
        n_kv = attn0.n_kv if self.is_llama else attn0.n_heads
        dtype = next(self.m.parameters()).dtype
        L = S0 + max_new_tokens
        cache = KVCache(n_layers, B, n_kv, L, attn0.head_dim,
                        prompt.device, dtype)
        out = torch.zeros(B, L, dtype=prompt.dtype, device=prompt.device)
        out[:, :S0] = prompt
        logits = self._forward_tokens(prompt, cache, 0)   # prefill
        nxt = logits.argmax(dim=-1, keepdim=True)
        out[:, S0] = nxt[:, 0]
        if max_new_tokens == 1:
            return out
        cur_tok = nxt.clone()
        # pos_dev = position the NEXT step writes its kv/token at
        pos_dev = torch.tensor([S0], dtype=torch.int64, device=prompt.device)
        steps = max_new_tokens - 1
        if not (use_graph and prompt.is_cuda):
            for _ in range(steps):
                self._decode_step_static(cur_tok, cache, pos_dev, out)
            return out
        # warm up twice on a side stream, then capture one step
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        did = 0
        with torch.cuda.stream(s):
            for _ in range(min(2, steps)):
                self._decode_step_static(cur_tok, cache, pos_dev, out)
                did += 1
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            # capture RECORDS without executing: state does not advance
            self._decode_step_static(cur_tok, cache, pos_dev, out)
        for _ in range(steps - did):
            g.replay()
        return out

