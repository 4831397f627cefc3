"""LoRA fine-tuning: low-rank adapters on the linear projections.

Mechanism: `torch.nn.utils.parametrize` on each target Linear's
`weight` — attribute access returns `W + (alpha/r) * B @ A`, which is
exactly what the models' attribute-style call sites
(`ops.linear(x, self.qkv.weight, ...)`) consume, so NO model code
changes and the hand-kernel compute path is untouched.  The base
weights are frozen; only A/B train (BucketedDDP and MasterAdamW
already filter on requires_grad, and the optimizer carries explicit
param-name identities so checkpoints stay aligned with frozen params
present).

Cost note: each weight access materializes W + BA once per
forward/backward per layer (a rank-r outer product + add — bandwidth,
not MFMA-bound).  The classic x@A@B factored form would avoid the
materialization but requires module-call sites; this repo's models
deliberately call ops with raw weights (docs/KERNELS.md), so the
parametrization form is the right trade until a fused LoRA epilogue
kernel exists (r3 candidate).

Usage:
    apply_lora(model, r=8, alpha=16)        # or model.lora_r config
    ... train (only adapters get grads) ...
    sd = lora_state_dict(model)             # adapters only, tiny
    merge_lora(model)                       # bake W += s*BA for serving
"""

from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.nn as nn
from torch.nn.utils import parametrize


class LoRAAdapter(nn.Module):
    """Parametrization: weight -> weight + (alpha/r) * B @ A."""

    def __init__(self, out_dim: int, in_dim: int, r: int, alpha: float,
                 device=None, dtype=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.A = nn.Parameter(torch.randn(r, in_dim, **kw) * 0.02)
        self.B = nn.Parameter(torch.zeros(out_dim, r, **kw))
        self.scale = float(alpha) / float(r)

    def forward(self, w: torch.Tensor) -> torch.Tensor:
        return w + (self.B @ self.A).to(w.dtype) * self.scale


def _is_target(name: str, module: nn.Module, targets) -> bool:
    if not isinstance(module, nn.Linear):
        return False
    leaf = name.rsplit(".", 1)[-1]
    if targets is not None:
        return leaf in targets
    return leaf not in ("lm_head",)       # default: all Linears but the head


def apply_lora(model: nn.Module, r: int = 8, alpha: float = 16.0,
               targets: Optional[set] = None) -> int:
    """Freeze the model and attach adapters; returns #adapted layers.
    `targets` is a set of leaf module names (e.g. {"qkv", "proj"});
    None adapts every Linear except the (often tied) lm_head."""
    n = 0
    for pname, p in model.named_parameters():
        p.requires_grad_(False)
    for name, mod in model.named_modules():
        if _is_target(name, mod, targets):
            parametrize.register_parametrization(
                mod, "weight",
                LoRAAdapter(mod.out_features, mod.in_features, r, alpha,
                            device=mod.weight.device,
                            dtype=mod.weight.dtype))
            n += 1
    # freezing ran BEFORE registration, so only A/B are trainable now;
    # make it explicit for readers and robustness
    for name, p in model.named_parameters():
        p.requires_grad_(".parametrizations.weight.0." in name)
    if n == 0:
        raise ValueError("apply_lora: no target Linear layers found")
    return n


@torch.no_grad()
def merge_lora(model: nn.Module) -> nn.Module:
    """Bake adapters into the base weights (serving form) and remove
    the parametrizations; all params become plain and trainable."""
    for mod in list(model.modules()):
        if parametrize.is_parametrized(mod, "weight"):
            parametrize.remove_parametrizations(mod, "weight",
                                                leave_parametrized=True)
    for p in model.parameters():
        p.requires_grad_(True)
    return model


def lora_state_dict(model: nn.Module) -> Dict[str, torch.Tensor]:
    """Adapter tensors only (A/B) — the portable fine-tune artifact."""
    return {k: v.detach().cpu() for k, v in model.state_dict().items()
            if ".parametrizations.weight.0." in k}


def load_lora_state_dict(model: nn.Module, sd: Dict[str, torch.Tensor]) -> None:
    own = {k: v for k, v in model.state_dict().items()
           if ".parametrizations.weight.0." in k}
    if set(own) != set(sd):
        raise KeyError(
            f"adapter mismatch: missing {sorted(set(own) - set(sd))[:3]}, "
            f"unexpected {sorted(set(sd) - set(own))[:3]}")
    with torch.no_grad():
        for k, dst in own.items():
            dst.copy_(sd[k].to(dst.device, dst.dtype))
