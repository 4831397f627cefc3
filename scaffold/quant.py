"""Int8 weight-only quantization — the EXPORT side.

Per-output-channel symmetric int8 for every 2-D weight (Linear /
embedding rows): `w_q = round(w / s)`, `s = max|w_row| / 127`, stored as
int8 + fp32 scales in one safetensors file (~2x smaller than bf16,
~4x smaller than fp32).  `load_quantized` dequantizes back into a model
(serving-quality: per-channel symmetric keeps relative error ~1/254 of
each row's range).

What this is NOT yet: a quantized COMPUTE path.  The llama decode step
is weight-read-bound (NOTES r2 serving ladder), so an int8 weight x
bf16 activation GEMV kernel would ~halve decode step time — that
kernel is a round-3 item (docs/KERNELS.md conventions apply); until it
exists, quantization buys artifact size, not speed, and this module
says so rather than pretending otherwise.
"""

from __future__ import annotations

from typing import Dict, Tuple

import torch

_QSUF = "::int8"
_SSUF = "::scale"


def quantize_tensor(w: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """(out, in) fp -> (int8 (out, in), fp32 scale (out, 1))."""
    wf = w.detach().float()
    s = wf.abs().amax(dim=1, keepdim=True).clamp(min=1e-12) / 127.0
    q = torch.clamp(torch.round(wf / s), -127, 127).to(torch.int8)
    return q, s


def dequantize_tensor(q: torch.Tensor, s: torch.Tensor) -> torch.Tensor:
    return q.float() * s


def quantize_state_dict(model: torch.nn.Module) -> Dict[str, torch.Tensor]:
    """2-D weights -> int8+scale pairs; everything else stored raw.
    Tied tensors are stored once (by first name seen)."""
    out: Dict[str, torch.Tensor] = {}
    seen = {}
    for name, t in model.state_dict().items():
        ptr = t.untyped_storage().data_ptr()
        if ptr in seen:
            out[name + "::tied"] = torch.tensor(
                [], dtype=torch.int8)  # marker; target name in metadata
            continue
        seen[ptr] = name
        if t.dim() == 2 and t.is_floating_point():
            q, s = quantize_tensor(t)
            out[name + _QSUF] = q.cpu()
            out[name + _SSUF] = s.cpu()
        else:
            out[name] = t.detach().cpu().contiguous()
    return out


def save_quantized(model: torch.nn.Module, path: str) -> None:
    import safetensors.torch as st

    sd = quantize_state_dict(model)
    # record tie targets in metadata (marker tensors carry no data)
    ties = {}
    seen = {}
    for name, t in model.state_dict().items():
        ptr = t.untyped_storage().data_ptr()
        if ptr in seen:
            ties[name] = seen[ptr]
        else:
            seen[ptr] = name
    st.save_file(sd, path, metadata={"tied": repr(ties)} if ties else {})


@torch.no_grad()
def load_quantized(model: torch.nn.Module, path: str) -> None:
    """Dequantize into `model` in place (dtype/device follow the
    destination)."""
    import ast

    import safetensors.torch as st
    from safetensors import safe_open

    sd = st.load_file(path)
    with safe_open(path, framework="pt", device="cpu") as f:
        ties = ast.literal_eval((f.metadata() or {}).get("tied", "{}"))

    def fetch(name: str) -> torch.Tensor:
        name = ties.get(name, name)
        if name + _QSUF in sd:
            return dequantize_tensor(sd[name + _QSUF], sd[name + _SSUF])
        return sd[name]

    for name, t in model.state_dict().items():
        t.copy_(fetch(name).to(t.device, t.dtype))
