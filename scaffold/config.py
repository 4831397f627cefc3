"""Typed hierarchical training configuration.

Capability C2 of the contract (BASELINE.json "config schema"): a typed
config tree with model/data/optim/dist/ckpt/log sections, loadable from a
JSON or TOML file with ``key.sub=value`` CLI overrides.  The schema is
deliberately isolated in this one module and versioned
(``SCHEMA_VERSION``) so a later reconciliation against the reference's
exact schema is a single-file change.

No YAML/hydra/omegaconf: plain dataclasses + json/tomli, which are the
only config-parsing facilities in this environment.
"""

from __future__ import annotations

import dataclasses
import hashlib
import json
import os
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

SCHEMA_VERSION = 1


@dataclass
class ModelConfig:
    name: str = "mlp"  # mlp | resnet50 | gpt2 | llama3_8b | llama (custom dims)
    # MLP
    in_dim: int = 256
    hidden_dim: int = 512
    out_dim: int = 16
    # Vision
    num_classes: int = 1000
    # Transformer (gpt2/llama); defaults overridden per named preset
    vocab_size: int = 50257
    n_layers: int = 12
    n_heads: int = 12
    n_kv_heads: Optional[int] = None  # GQA; None -> n_heads
    d_model: int = 768
    d_ff: Optional[int] = None  # None -> 4*d_model (gpt2) / preset (llama)
    max_seq_len: int = 1024
    rope_theta: float = 500000.0
    # RoPE frequency scaling for long-context checkpoints:
    # "none" | "linear" (divide frequencies by rope_factor) | "llama3"
    # (the Llama-3.1 wavelength-dependent rule; matches transformers)
    rope_scaling: str = "none"
    rope_factor: float = 8.0
    rope_low_freq_factor: float = 1.0
    rope_high_freq_factor: float = 4.0
    rope_orig_max_pos: int = 8192
    norm_eps: float = 1e-5
    tie_embeddings: bool = True
    # gpt2: replace every block's dense FFN with a top-1-routed MoE of
    # N local experts (0 = dense).  Expert-parallel sharding of the same
    # layer across ranks is scaffold.dist.ep.ExpertParallelMoE
    n_experts: int = 0
    # experts per token (Mixtral-style weighted mixing when > 1)
    moe_top_k: int = 1
    # LoRA fine-tuning: rank > 0 freezes the model and attaches
    # low-rank adapters to every Linear except lm_head (scaffold/lora.py)
    lora_r: int = 0
    lora_alpha: float = 16.0
    # recompute each transformer block in backward instead of storing
    # its activations (trades ~1 extra block fwd for O(1) activation
    # memory per block); LM families only, no-op in eval
    act_ckpt: bool = False


@dataclass
class DataConfig:
    kind: str = "synthetic_image"  # synthetic_image | synthetic_tokens | synthetic_vector
    batch_size: int = 32  # per-rank batch size
    image_size: int = 224
    channels: int = 3
    seq_len: int = 1024
    num_classes: int = 1000
    vector_dim: int = 256
    active_vocab: int = 0  # LM data: restrict token ids to [0,N) for learnable signal
    path: str = ""  # token_file: flat binary of token ids (uint16/uint32 by vocab)
    seed: int = 1234


@dataclass
class OptimConfig:
    name: str = "adamw"
    lr: float = 1e-3
    betas: List[float] = field(default_factory=lambda: [0.9, 0.95])
    eps: float = 1e-8
    weight_decay: float = 0.01
    grad_clip: float = 0.0  # global-norm clip; 0 = off
    fused: bool = True  # use the HIP fused multi-tensor kernel on GPU
    schedule: str = "const"  # const | cosine | linear
    warmup_steps: int = 0
    min_lr_ratio: float = 0.1  # floor for cosine/linear decay


@dataclass
class DistConfig:
    backend: str = "auto"  # auto | nccl (=RCCL on ROCm) | gloo
    bucket_cap_mb: float = 25
    first_bucket_mb: float = 4  # small first bucket fires early in backward
    overlap: bool = True  # all-reduce overlapped with backward (C4)
    min_nchannels: int = 7  # RCCL channels >= xGMI links per GPU
    comm: str = "auto"  # auto | rccl (direct C++ bucket path) | torch
    timeout_s: int = 600
    # ZeRO-1: shard optimizer state (fp32 master + moments) across the
    # DP group; grads stay fully all-reduced (bucket engine unchanged),
    # each rank steps only its owned params, then owners broadcast the
    # updated working params.  Cuts optimizer memory by ~world_size
    # (Llama-8B: 96 GB -> 12 GB/rank at 8 GPUs).
    zero1: bool = False
    # expert parallelism for the MoE family (model.n_experts > 0):
    # shard the experts across the WORLD (n_experts % world == 0),
    # routed tokens exchanged by all_to_all; backbone/gate stay
    # data-parallel.  Mutually exclusive with zero1 (expert params are
    # rank-local, so a name-keyed optimizer partition would collide).
    ep_moe: bool = False


@dataclass
class CkptConfig:
    dir: str = ""
    interval: int = 0  # steps between checkpoints; 0 = only final
    resume: str = ""  # path to a step_NNNNNN dir to resume from
    # initial weights for fine-tuning (NOT a resume: optimizer/RNG/data
    # start fresh): a .safetensors file (scaffold.convert.save_model
    # layout), an int8 quantized export, or an HF-layout directory
    init_weights: str = ""
    save_final: bool = False
    async_d2h: bool = True  # pinned hipMemcpyAsync staging on a side stream (C6)


@dataclass
class LogConfig:
    interval: int = 10
    metrics_file: str = ""  # rank-0 JSONL; "" = disabled
    profile_steps: int = 0  # wrap N steps in torch.profiler if >0
    eval_interval: int = 0  # run evaluate() every N steps; 0 = off
    eval_steps: int = 10


@dataclass
class TrainConfig:
    model: ModelConfig = field(default_factory=ModelConfig)
    data: DataConfig = field(default_factory=DataConfig)
    optim: OptimConfig = field(default_factory=OptimConfig)
    dist: DistConfig = field(default_factory=DistConfig)
    ckpt: CkptConfig = field(default_factory=CkptConfig)
    log: LogConfig = field(default_factory=LogConfig)
    steps: int = 100
    grad_accum: int = 1
    cudnn_benchmark: bool = True  # MIOpen exhaustive find (cached in miopen_db)
    hip_graph: bool = False  # capture the train step in a hipGraph and replay
    seed: int = 0
    dtype: str = "bf16"  # compute dtype: bf16 | fp32
    device: str = "auto"  # auto | cuda | cpu
    schema_version: int = SCHEMA_VERSION


_SECTION_TYPES = {
    "model": ModelConfig,
    "data": DataConfig,
    "optim": OptimConfig,
    "dist": DistConfig,
    "ckpt": CkptConfig,
    "log": LogConfig,
}


def _coerce(value: str, target_type: Any) -> Any:
    """Coerce a CLI string override to the dataclass field's type."""
    if target_type is bool or isinstance(target_type, bool):
        if isinstance(value, bool):
            return value
        return str(value).lower() in ("1", "true", "yes", "on")
    if target_type is int:
        return int(value)
    if target_type is float:
        return float(value)
    if target_type in (str,):
        return str(value)
    # Optional[...] / List[...] -> try json
    try:
        return json.loads(value)
    except (json.JSONDecodeError, TypeError):
        return value


def _field_type(dc: Any, name: str) -> Any:
    for f in dataclasses.fields(dc):
        if f.name == name:
            t = f.type
            if isinstance(t, str):
                # from __future__ annotations: resolve basic names
                t = {"int": int, "float": float, "str": str, "bool": bool}.get(
                    t.replace("Optional[", "").replace("]", ""), str
                )
            return t
    raise KeyError(f"unknown config field {name!r} on {type(dc).__name__}")


def config_from_dict(d: Dict[str, Any]) -> TrainConfig:
    cfg = TrainConfig()
    for key, val in d.items():
        if key in _SECTION_TYPES and isinstance(val, dict):
            section = getattr(cfg, key)
            for k, v in val.items():
                if not hasattr(section, k):
                    raise KeyError(f"unknown config field {key}.{k}")
                setattr(section, k, v)
        else:
            if not hasattr(cfg, key):
                raise KeyError(f"unknown top-level config field {key}")
            setattr(cfg, key, val)
    return cfg


def config_to_dict(cfg: TrainConfig) -> Dict[str, Any]:
    return dataclasses.asdict(cfg)


def apply_overrides(cfg: TrainConfig, overrides: List[str]) -> TrainConfig:
    """Apply ``section.key=value`` / ``key=value`` CLI overrides in place."""
    for ov in overrides:
        if "=" not in ov:
            raise ValueError(f"override {ov!r} is not key=value")
        path, value = ov.split("=", 1)
        parts = path.split(".")
        if len(parts) == 1:
            ft = _field_type(cfg, parts[0])
            setattr(cfg, parts[0], _coerce(value, ft))
        elif len(parts) == 2 and parts[0] in _SECTION_TYPES:
            section = getattr(cfg, parts[0])
            ft = _field_type(section, parts[1])
            setattr(section, parts[1], _coerce(value, ft))
        else:
            raise KeyError(f"unknown config path {path!r}")
    return cfg


def load_config(path: Optional[str] = None, overrides: Optional[List[str]] = None) -> TrainConfig:
    """Load a TrainConfig from a JSON/TOML file plus CLI overrides."""
    d: Dict[str, Any] = {}
    if path:
        if not os.path.exists(path):
            raise FileNotFoundError(path)
        if path.endswith(".toml"):
            import tomli

            with open(path, "rb") as fh:
                d = tomli.load(fh)
        else:
            with open(path) as fh:
                d = json.load(fh)
    cfg = config_from_dict(d)
    if overrides:
        apply_overrides(cfg, overrides)
    _apply_model_preset(cfg)
    return cfg


def _apply_model_preset(cfg: TrainConfig) -> None:
    """Fill model dims for named presets unless the user overrode them."""
    m = cfg.model
    if m.name == "gpt2":
        if m.d_ff is None:
            m.d_ff = 4 * m.d_model
    elif m.name in ("llama3_8b", "llama"):
        if m.name == "llama3_8b":
            m.vocab_size = 128256
            m.n_layers = 32
            m.n_heads = 32
            m.n_kv_heads = 8
            m.d_model = 4096
            m.d_ff = 14336
            m.max_seq_len = 4096 if m.max_seq_len == 1024 else m.max_seq_len
        if m.d_ff is None:
            m.d_ff = int(8 * m.d_model / 3 + 255) // 256 * 256
        if m.n_kv_heads is None:
            m.n_kv_heads = m.n_heads


def config_hash(cfg: TrainConfig) -> str:
    """Stable hash of the config, stored in checkpoint manifests."""
    blob = json.dumps(config_to_dict(cfg), sort_keys=True).encode()
    return hashlib.sha256(blob).hexdigest()[:16]
