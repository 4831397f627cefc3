"""MI355X-native distributed training scaffold.

A from-scratch rebuild of the FRL-Distributed-ML-Scaffold capability
surface designed for MI355X: PyTorch-ROCm autograd +
hand-written CDNA4 HIP kernels (csrc/) + RCCL over xGMI.
"""

import os as _os

# MIOpen find/tuning DB + compiled-kernel cache pinned INSIDE the repo:
# torch.backends.cudnn.benchmark=True triggers MIOpen's exhaustive conv
# algo search (~3 min of warmup for ResNet-50 on a fresh box — measured
# via rocprof, profiles/resnet_kstats notes); persisting the DB in-tree
# means every fresh machine reuses the tuned choices instantly.  Must be
# set before MIOpen's first use; harmless on CPU.
_REPO_ROOT = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
_MIOPEN_DB = _os.path.join(_REPO_ROOT, "miopen_db")
_os.makedirs(_MIOPEN_DB, exist_ok=True)
_os.environ.setdefault("MIOPEN_USER_DB_PATH", _MIOPEN_DB)
_os.environ.setdefault("MIOPEN_CUSTOM_CACHE_DIR", _MIOPEN_DB)

from .config import TrainConfig, load_config, config_hash
from .runtime import init_distributed, DistContext
from .trainer import Trainer

__version__ = "0.2.0"

# lazy attributes (PEP 562): the serving / interop / fine-tune surface
# pulls optional deps (fastapi, tokenizers) or large modules only on use
_LAZY = {
    "Generator": ("scaffold.generate", "Generator"),
    "KVCache": ("scaffold.generate", "KVCache"),
    "InferenceWorker": ("scaffold.serve", "InferenceWorker"),
    "ContinuousBatcher": ("scaffold.serve", "ContinuousBatcher"),
    "create_app": ("scaffold.serve", "create_app"),
    "load_hf": ("scaffold.convert", "load_hf"),
    "load_hf_dir": ("scaffold.convert", "load_hf_dir"),
    "save_model": ("scaffold.convert", "save_model"),
    "load_model": ("scaffold.convert", "load_model"),
    "apply_lora": ("scaffold.lora", "apply_lora"),
    "merge_lora": ("scaffold.lora", "merge_lora"),
    "save_quantized": ("scaffold.quant", "save_quantized"),
    "load_quantized": ("scaffold.quant", "load_quantized"),
    "run_elastic": ("scaffold.launch", "run_elastic"),
    "shard_llama_tp": ("scaffold.dist.tp", "shard_llama_tp"),
    "TpContinuousBatcher": ("scaffold.serve_tp", "TpContinuousBatcher"),
    "run_follower": ("scaffold.serve_tp", "run_follower"),
    "ExpertParallelMoE": ("scaffold.dist.ep", "ExpertParallelMoE"),
}


def __getattr__(name):
    if name in _LAZY:
        import importlib

        mod, attr = _LAZY[name]
        return getattr(importlib.import_module(mod), attr)
    raise AttributeError(f"module 'scaffold' has no attribute {name!r}")


__all__ = ["TrainConfig", "load_config", "config_hash", "init_distributed",
           "DistContext", "Trainer", *sorted(_LAZY)]
