"""Hugging Face checkpoint interop: map `transformers`-layout GPT-2 /
Llama state dicts onto the scaffold's models.

Two conventions differ from HF and are translated here, not in the
model code:

- GPT-2 stores its projections as Conv1D ([in, out]) — transposed
  relative to nn.Linear, and packs qkv in one [in, 3*d] matrix, which
  matches our fused `attn.qkv` after the transpose.
- Llama RoPE: HF uses the half-split ("rotate_half") layout, this repo
  uses the original interleaved-pair layout (ops._rope_eager — pairs
  (x[2i], x[2i+1])).  The two are related by a fixed row permutation of
  W_q/W_k (the inverse of the permutation the official meta->HF
  converter applies), so converted weights give identical logits.

`load_hf(model, state_dict)` detects the family from the model class
and copies in place; logit parity against `transformers` itself is
proven by tests/integration/test_hf_parity.py.
"""

from __future__ import annotations

from typing import Dict

import torch

from .models.gpt2 import GPT2
from .models.llama import Llama


def _strip_prefix(sd: Dict[str, torch.Tensor], prefix: str) -> Dict[str, torch.Tensor]:
    if any(k.startswith(prefix) for k in sd):
        return {k[len(prefix):] if k.startswith(prefix) else k: v
                for k, v in sd.items()}
    return dict(sd)


def _unpermute_rope(w: torch.Tensor, n_heads: int) -> torch.Tensor:
    """HF rotate_half row layout -> interleaved-pair row layout.

    The meta->HF converter reshapes each head's rows as
    (head_dim/2, 2) -> transpose; invert that: (2, head_dim/2) ->
    transpose back."""
    out_dim, in_dim = w.shape
    hd = out_dim // n_heads
    return (w.view(n_heads, 2, hd // 2, in_dim)
             .transpose(1, 2).reshape(out_dim, in_dim))


def from_hf_gpt2(sd: Dict[str, torch.Tensor], n_layers: int) -> Dict[str, torch.Tensor]:
    """transformers GPT2LMHeadModel state dict -> scaffold GPT2 names."""
    sd = _strip_prefix(sd, "transformer.")
    out = {
        "wte.weight": sd["wte.weight"],
        "wpe.weight": sd["wpe.weight"],
        "ln_f.weight": sd["ln_f.weight"],
        "ln_f.bias": sd["ln_f.bias"],
        "lm_head.weight": sd.get("lm_head.weight", sd["wte.weight"]),
    }
    for i in range(n_layers):
        h, b = f"h.{i}.", f"blocks.{i}."
        out[b + "ln1.weight"] = sd[h + "ln_1.weight"]
        out[b + "ln1.bias"] = sd[h + "ln_1.bias"]
        # Conv1D stores [in, out] -> transpose to nn.Linear [out, in]
        out[b + "attn.qkv.weight"] = sd[h + "attn.c_attn.weight"].t().contiguous()
        out[b + "attn.qkv.bias"] = sd[h + "attn.c_attn.bias"]
        out[b + "attn.proj.weight"] = sd[h + "attn.c_proj.weight"].t().contiguous()
        out[b + "attn.proj.bias"] = sd[h + "attn.c_proj.bias"]
        out[b + "ln2.weight"] = sd[h + "ln_2.weight"]
        out[b + "ln2.bias"] = sd[h + "ln_2.bias"]
        out[b + "fc.weight"] = sd[h + "mlp.c_fc.weight"].t().contiguous()
        out[b + "fc.bias"] = sd[h + "mlp.c_fc.bias"]
        out[b + "proj.weight"] = sd[h + "mlp.c_proj.weight"].t().contiguous()
        out[b + "proj.bias"] = sd[h + "mlp.c_proj.bias"]
    return out


def from_hf_llama(sd: Dict[str, torch.Tensor], n_layers: int, n_heads: int,
                  n_kv_heads: int) -> Dict[str, torch.Tensor]:
    """transformers LlamaForCausalLM state dict -> scaffold Llama names."""
    lm_head = sd.get("lm_head.weight")
    sd = _strip_prefix(sd, "model.")
    out = {
        "embed.weight": sd["embed_tokens.weight"],
        "norm_f.weight": sd["norm.weight"],
        "lm_head.weight": lm_head if lm_head is not None
        else sd["embed_tokens.weight"],
    }
    for i in range(n_layers):
        h, b = f"layers.{i}.", f"blocks.{i}."
        out[b + "attn_norm.weight"] = sd[h + "input_layernorm.weight"]
        out[b + "mlp_norm.weight"] = sd[h + "post_attention_layernorm.weight"]
        q = _unpermute_rope(sd[h + "self_attn.q_proj.weight"], n_heads)
        k = _unpermute_rope(sd[h + "self_attn.k_proj.weight"], n_kv_heads)
        v = sd[h + "self_attn.v_proj.weight"]
        out[b + "attn.wqkv.weight"] = torch.cat([q, k, v], dim=0)
        out[b + "attn.wo.weight"] = sd[h + "self_attn.o_proj.weight"]
        out[b + "w_gateup.weight"] = torch.cat(
            [sd[h + "mlp.gate_proj.weight"], sd[h + "mlp.up_proj.weight"]], dim=0)
        out[b + "w_down.weight"] = sd[h + "mlp.down_proj.weight"]
    return out


@torch.no_grad()
def load_hf(model: torch.nn.Module, sd: Dict[str, torch.Tensor]) -> None:
    """Copy a transformers state dict into a scaffold model in place.
    Family is detected from the model class; dtype/device follow the
    destination parameters."""
    if isinstance(model, GPT2):
        mapped = from_hf_gpt2(sd, len(model.blocks))
    elif isinstance(model, Llama):
        a0 = model.blocks[0].attn
        if any("mlp.experts.gate_up_proj" in k for k in sd):
            mapped = from_hf_mixtral(sd, len(model.blocks), a0.n_heads,
                                     a0.n_kv)
        else:
            mapped = from_hf_llama(sd, len(model.blocks), a0.n_heads,
                                   a0.n_kv)
    else:
        raise TypeError(f"load_hf: unsupported model {type(model).__name__}")
    own = dict(model.named_parameters())
    missing = [k for k in own if k not in mapped]
    if missing:
        raise KeyError(f"load_hf: no source for {missing}")
    for name, src in mapped.items():
        if name in own:
            p = own[name]
            if p.shape != src.shape:
                raise ValueError(
                    f"load_hf: {name} shape {tuple(src.shape)} != "
                    f"{tuple(p.shape)}")
            p.copy_(src.to(p.device, p.dtype))


def load_hf_dir(model: torch.nn.Module, path: str) -> None:
    """Load from a local HF-layout directory (*.safetensors shards or
    pytorch_model.bin)."""
    import glob
    import os

    sd: Dict[str, torch.Tensor] = {}
    shards = sorted(glob.glob(os.path.join(path, "*.safetensors")))
    if shards:
        import safetensors.torch as st
        for f in shards:
            sd.update(st.load_file(f))
    else:
        binp = os.path.join(path, "pytorch_model.bin")
        sd = torch.load(binp, map_location="cpu", weights_only=True)
    load_hf(model, sd)


def _permute_rope(w: torch.Tensor, n_heads: int) -> torch.Tensor:
    """Interleaved-pair row layout -> HF rotate_half (inverse of
    _unpermute_rope)."""
    out_dim, in_dim = w.shape
    hd = out_dim // n_heads
    return (w.view(n_heads, hd // 2, 2, in_dim)
             .transpose(1, 2).reshape(out_dim, in_dim))


@torch.no_grad()
def to_hf_gpt2(model: GPT2) -> Dict[str, torch.Tensor]:
    """scaffold GPT2 -> transformers GPT2LMHeadModel state dict
    (inverse of from_hf_gpt2; roundtrip-tested)."""
    sd = {k: v.detach().cpu() for k, v in model.state_dict().items()}
    out = {
        "transformer.wte.weight": sd["wte.weight"],
        "transformer.wpe.weight": sd["wpe.weight"],
        "transformer.ln_f.weight": sd["ln_f.weight"],
        "transformer.ln_f.bias": sd["ln_f.bias"],
        "lm_head.weight": sd["lm_head.weight"],
    }
    for i in range(len(model.blocks)):
        h, b = f"transformer.h.{i}.", f"blocks.{i}."
        out[h + "ln_1.weight"] = sd[b + "ln1.weight"]
        out[h + "ln_1.bias"] = sd[b + "ln1.bias"]
        out[h + "attn.c_attn.weight"] = sd[b + "attn.qkv.weight"].t().contiguous()
        out[h + "attn.c_attn.bias"] = sd[b + "attn.qkv.bias"]
        out[h + "attn.c_proj.weight"] = sd[b + "attn.proj.weight"].t().contiguous()
        out[h + "attn.c_proj.bias"] = sd[b + "attn.proj.bias"]
        out[h + "ln_2.weight"] = sd[b + "ln2.weight"]
        out[h + "ln_2.bias"] = sd[b + "ln2.bias"]
        out[h + "mlp.c_fc.weight"] = sd[b + "fc.weight"].t().contiguous()
        out[h + "mlp.c_fc.bias"] = sd[b + "fc.bias"]
        out[h + "mlp.c_proj.weight"] = sd[b + "proj.weight"].t().contiguous()
        out[h + "mlp.c_proj.bias"] = sd[b + "proj.bias"]
    return out


@torch.no_grad()
def to_hf_llama(model: Llama) -> Dict[str, torch.Tensor]:
    """scaffold Llama -> transformers LlamaForCausalLM state dict."""
    sd = {k: v.detach().cpu() for k, v in model.state_dict().items()}
    a0 = model.blocks[0].attn
    nh, nkv, hd = a0.n_heads, a0.n_kv, a0.head_dim
    out = {
        "model.embed_tokens.weight": sd["embed.weight"],
        "model.norm.weight": sd["norm_f.weight"],
        "lm_head.weight": sd["lm_head.weight"],
    }
    for i in range(len(model.blocks)):
        h, b = f"model.layers.{i}.", f"blocks.{i}."
        out[h + "input_layernorm.weight"] = sd[b + "attn_norm.weight"]
        out[h + "post_attention_layernorm.weight"] = sd[b + "mlp_norm.weight"]
        wqkv = sd[b + "attn.wqkv.weight"]
        q, k, v = torch.split(wqkv, [nh * hd, nkv * hd, nkv * hd], dim=0)
        out[h + "self_attn.q_proj.weight"] = _permute_rope(q, nh)
        out[h + "self_attn.k_proj.weight"] = _permute_rope(k, nkv)
        out[h + "self_attn.v_proj.weight"] = v.contiguous()
        out[h + "self_attn.o_proj.weight"] = sd[b + "attn.wo.weight"]
        gu = sd[b + "w_gateup.weight"]
        f2 = gu.shape[0] // 2
        out[h + "mlp.gate_proj.weight"] = gu[:f2].contiguous()
        out[h + "mlp.up_proj.weight"] = gu[f2:].contiguous()
        out[h + "mlp.down_proj.weight"] = sd[b + "w_down.weight"]
    return out


def save_model(model: torch.nn.Module, path: str) -> None:
    """Single-file safetensors weight export (serving deployments —
    trainer state stays with the sharded checkpoint engine).  Tied
    tensors (GPT-2 wte/lm_head) are stored once and re-tied on load."""
    import safetensors.torch as st

    sd = {}
    seen = {}
    tied = {}
    for k, v in model.state_dict().items():
        ptr = v.untyped_storage().data_ptr()
        if ptr in seen:
            tied[k] = seen[ptr]
        else:
            seen[ptr] = k
            sd[k] = v.detach().cpu().contiguous()
    st.save_file(sd, path, metadata={"tied": repr(tied)} if tied else {})


def load_model(model: torch.nn.Module, path: str) -> None:
    """Inverse of save_model; dtype/device follow the destination."""
    import ast

    import safetensors.torch as st
    from safetensors import safe_open

    with safe_open(path, framework="pt", device="cpu") as f:
        meta = f.metadata() or {}
    tied = ast.literal_eval(meta.get("tied", "{}"))
    sd = st.load_file(path)
    for k, src in tied.items():
        sd[k] = sd[src]
    own = model.state_dict()
    missing = [k for k in own if k not in sd]
    if missing:
        raise KeyError(f"load_model: missing {missing}")
    with torch.no_grad():
        for k, t in own.items():
            t.copy_(sd[k].to(t.device, t.dtype))


def from_hf_mixtral(sd: Dict[str, torch.Tensor], n_layers: int, n_heads: int,
                    n_kv_heads: int) -> Dict[str, torch.Tensor]:
    """transformers MixtralForCausalLM -> scaffold Llama-MoE names.

    Attention maps like Llama (rotate-half unpermute).  transformers'
    current Mixtral stores the experts FUSED: gate_up_proj (E, 2F, d)
    whose rows chunk to gate|up — the exact layout of our packed
    w_up — and down_proj (E, d, F); the router (full-logit softmax ->
    top-k -> renormalize) matches this repo's routing bit-for-bit, so
    parity holds end to end (tested)."""
    lm_head = sd.get("lm_head.weight")
    sd = _strip_prefix(sd, "model.")
    out = {
        "embed.weight": sd["embed_tokens.weight"],
        "norm_f.weight": sd["norm.weight"],
        "lm_head.weight": lm_head if lm_head is not None
        else sd["embed_tokens.weight"],
    }
    for i in range(n_layers):
        h, b = f"layers.{i}.", f"blocks.{i}."
        out[b + "attn_norm.weight"] = sd[h + "input_layernorm.weight"]
        out[b + "mlp_norm.weight"] = sd[h + "post_attention_layernorm.weight"]
        q = _unpermute_rope(sd[h + "self_attn.q_proj.weight"], n_heads)
        k = _unpermute_rope(sd[h + "self_attn.k_proj.weight"], n_kv_heads)
        v = sd[h + "self_attn.v_proj.weight"]
        out[b + "attn.wqkv.weight"] = torch.cat([q, k, v], dim=0)
        out[b + "attn.wo.weight"] = sd[h + "self_attn.o_proj.weight"]
        out[b + "moe.gate.weight"] = sd[h + "mlp.gate.weight"]
        gu = sd[h + "mlp.experts.gate_up_proj"]      # (E, 2F, d)
        dn = sd[h + "mlp.experts.down_proj"]         # (E, d, F)
        for e in range(gu.shape[0]):
            out[b + f"moe.w_up.{e}"] = gu[e].contiguous()
            out[b + f"moe.w_down.{e}"] = dn[e].contiguous()
    return out


@torch.no_grad()
def to_hf_mixtral(model: Llama) -> Dict[str, torch.Tensor]:
    """scaffold Llama-MoE -> transformers MixtralForCausalLM state dict
    (fused expert tensors; inverse of from_hf_mixtral)."""
    if getattr(model.blocks[0], "moe", None) is None:
        raise TypeError("to_hf_mixtral needs a Llama-MoE model "
                        "(model.n_experts > 0)")
    sd = {k: v.detach().cpu() for k, v in model.state_dict().items()}
    a0 = model.blocks[0].attn
    nh, nkv, hd = a0.n_heads, a0.n_kv, a0.head_dim
    out = {
        "model.embed_tokens.weight": sd["embed.weight"],
        "model.norm.weight": sd["norm_f.weight"],
        "lm_head.weight": sd["lm_head.weight"],
    }
    for i, blk in enumerate(model.blocks):
        h, b = f"model.layers.{i}.", f"blocks.{i}."
        out[h + "input_layernorm.weight"] = sd[b + "attn_norm.weight"]
        out[h + "post_attention_layernorm.weight"] = sd[b + "mlp_norm.weight"]
        wqkv = sd[b + "attn.wqkv.weight"]
        q, k, v = torch.split(wqkv, [nh * hd, nkv * hd, nkv * hd], dim=0)
        out[h + "self_attn.q_proj.weight"] = _permute_rope(q, nh)
        out[h + "self_attn.k_proj.weight"] = _permute_rope(k, nkv)
        out[h + "self_attn.v_proj.weight"] = v.contiguous()
        out[h + "self_attn.o_proj.weight"] = sd[b + "attn.wo.weight"]
        out[h + "mlp.gate.weight"] = sd[b + "moe.gate.weight"]
        ups = torch.stack([sd[b + f"moe.w_up.{e}"]
                           for e in range(blk.moe.n_experts)])
        downs = torch.stack([sd[b + f"moe.w_down.{e}"]
                             for e in range(blk.moe.n_experts)])
        out[h + "mlp.experts.gate_up_proj"] = ups
        out[h + "mlp.experts.down_proj"] = downs
    return out
