"""Model zoo (contract C9): MLP, ResNet-50, GPT-2-small, Llama-3-8B."""

from __future__ import annotations

import torch

from .mlp import MLP
from .resnet import ResNet50
from .gpt2 import GPT2
from .llama import Llama

__all__ = ["MLP", "ResNet50", "GPT2", "Llama", "build_model"]


def build_model(cfg) -> torch.nn.Module:
    name = cfg.model.name
    if name == "mlp":
        return MLP(cfg)
    if name == "resnet50":
        return ResNet50(cfg)
    if name == "gpt2":
        return GPT2(cfg)
    if name in ("llama", "llama3_8b"):
        return Llama(cfg)
    raise ValueError(f"unknown model {name!r}")
