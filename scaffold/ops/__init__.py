"""Hot ops (contract C5): fused AdamW, LayerNorm, softmax-cross-entropy,
attention, RMSNorm/RoPE.

Dispatch rule (C8 — no multi-backend dispatch on the compute path):
  * tensors on a CUDA (ROCm) device -> the hand-written CDNA4 HIP kernels
    from the in-tree extension, REQUIRED (loud error if the .so is
    missing) so GPU runs can never silently fall back to eager;
  * tensors on CPU -> plain PyTorch eager, the no-GPU test path (C10)
    and the fp32 numerics reference the kernel tests compare against.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from . import backend
from .backend import require_ext, has_ext, ext  # re-export

__all__ = [
    "layer_norm",
    "rms_norm",
    "rope_apply",
    "softmax_cross_entropy",
    "attention",
    "linear",
    "linear_gelu",
    "linear_residual",
    "swiglu",
    "swiglu_packed",
    "lm_head_ce",
    "fused_adamw_step",
    "batch_norm_act",
    "require_ext",
    "has_ext",
]


# ---------------------------------------------------------------- LayerNorm
class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        m = require_ext()
        x2 = x.contiguous().view(-1, x.shape[-1])
        y, mean, rstd = m.layernorm_fwd(x2, weight, bias, eps)
        ctx.save_for_backward(x2, weight, mean, rstd)
        ctx.x_shape = x.shape
        return y.view(x.shape)

    @staticmethod
    def backward(ctx, dy):
        x2, weight, mean, rstd = ctx.saved_tensors
        dy2 = dy.contiguous().view(-1, dy.shape[-1])
        m = require_ext()
        dx, dw, db = m.layernorm_bwd(dy2, x2, weight, mean, rstd)
        return dx.view(ctx.x_shape), dw, db, None


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
               eps: float = 1e-5) -> torch.Tensor:
    if x.is_cuda:
        return _LayerNorm.apply(x, weight, bias, eps)
    return F.layer_norm(x, (x.shape[-1],), weight, bias, eps)


# ----------------------------------------------------------------- RMSNorm
class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        m = require_ext()
        x2 = x.contiguous().view(-1, x.shape[-1])
        y, rstd = m.rmsnorm_fwd(x2, weight, eps)
        ctx.save_for_backward(x2, weight, rstd)
        ctx.x_shape = x.shape
        return y.view(x.shape)

    @staticmethod
    def backward(ctx, dy):
        x2, weight, rstd = ctx.saved_tensors
        dy2 = dy.contiguous().view(-1, dy.shape[-1])
        m = require_ext()
        dx, dw = m.rmsnorm_bwd(dy2, x2, weight, rstd)
        return dx.view(ctx.x_shape), dw, None


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if x.is_cuda:
        return _RMSNorm.apply(x, weight, eps)
    xf = x.float()
    y = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (y * weight.float()).to(x.dtype)


# -------------------------------------------------------------------- RoPE
class _Rope(torch.autograd.Function):
    """Rotary embedding on (B, H, S, D); cos/sin are (S, D/2) fp32."""

    @staticmethod
    def forward(ctx, x, cos, sin):
        m = require_ext()
        cos = cos.float().contiguous()  # tables stay fp32 even if model
        sin = sin.float().contiguous()  # buffers were cast to bf16
        y = m.rope_apply(x, cos, sin, False)
        ctx.save_for_backward(cos, sin)
        return y

    @staticmethod
    def backward(ctx, dy):
        cos, sin = ctx.saved_tensors
        m = require_ext()
        dx = m.rope_apply(dy, cos, sin, True)  # inverse rotation
        return dx, None, None


def _rope_eager(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                inverse: bool = False) -> torch.Tensor:
    # x: (B,H,S,D); pairs (x[2i], x[2i+1]) rotated by angle theta_i at pos s.
    xf = x.float()
    x1 = xf[..., 0::2]
    x2 = xf[..., 1::2]
    c = cos.view(1, 1, *cos.shape)
    s = sin.view(1, 1, *sin.shape)
    if inverse:
        s = -s
    o1 = x1 * c - x2 * s
    o2 = x1 * s + x2 * c
    out = torch.empty_like(xf)
    out[..., 0::2] = o1
    out[..., 1::2] = o2
    return out.to(x.dtype)


def rope_apply(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        return _Rope.apply(x, cos, sin)
    if x.requires_grad:
        # eager autograd handles the backward symbolically
        return _rope_eager(x, cos, sin)
    return _rope_eager(x, cos, sin)


def rope_tables(seq_len: int, dim: int, theta: float, device, dtype=torch.float32,
                scaling: str = "none", factor: float = 8.0,
                low_freq_factor: float = 1.0, high_freq_factor: float = 4.0,
                orig_max_pos: int = 8192
                ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Host-precomputed cos/sin tables (S, D/2) — Appendix B guidance:
    never compute trig per element on device.

    scaling="linear" divides every frequency by `factor`;
    scaling="llama3" applies the Llama-3.1 wavelength-dependent rule
    (long wavelengths scaled by 1/factor, short kept, smooth ramp in
    between) — parity with transformers' implementation is tested in
    tests/integration/test_hf_parity.py."""
    inv = 1.0 / (theta ** (torch.arange(0, dim, 2, dtype=torch.float32) / dim))
    if scaling == "linear":
        inv = inv / factor
    elif scaling == "llama3":
        import math as _math

        low_wl = orig_max_pos / low_freq_factor
        high_wl = orig_max_pos / high_freq_factor
        wavelen = 2 * _math.pi / inv
        inv_scaled = torch.where(wavelen > low_wl, inv / factor, inv)
        smooth = ((orig_max_pos / wavelen - low_freq_factor)
                  / (high_freq_factor - low_freq_factor))
        mid = (1 - smooth) * inv / factor + smooth * inv
        in_ramp = (wavelen <= low_wl) & (wavelen >= high_wl)
        inv = torch.where(in_ramp, mid, inv_scaled)
    elif scaling != "none":
        raise ValueError(f"unknown rope scaling {scaling!r}")
    t = torch.arange(seq_len, dtype=torch.float32)
    ang = torch.outer(t, inv)
    return ang.cos().to(device=device, dtype=dtype), ang.sin().to(device=device, dtype=dtype)


# ------------------------------------------------- softmax cross-entropy
class _SoftmaxXent(torch.autograd.Function):
    """Fused log-softmax + NLL; saves softmax stats, fuses dlogits in bwd."""

    @staticmethod
    def forward(ctx, logits, targets):
        m = require_ext()
        logits2 = logits.contiguous().view(-1, logits.shape[-1])
        t = targets.contiguous().view(-1)
        loss, lse = m.softmax_xent_fwd(logits2, t)
        ctx.save_for_backward(logits2, t, lse)
        ctx.logits_shape = logits.shape
        return loss.mean()  # mean reduction; bwd kernel folds the 1/R

    @staticmethod
    def backward(ctx, dloss):
        logits2, t, lse = ctx.saved_tensors
        m = require_ext()
        dlogits = m.softmax_xent_bwd(logits2, t, lse, float(dloss))
        return dlogits.view(ctx.logits_shape), None


def softmax_cross_entropy(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    """Mean cross-entropy over all positions. logits (..., V), targets (...)."""
    if logits.is_cuda:
        return _SoftmaxXent.apply(logits, targets)
    return F.cross_entropy(
        logits.view(-1, logits.shape[-1]).float(), targets.view(-1)
    )


# --------------------------------------------------------------- attention
class _Attention(torch.autograd.Function):
    """Flash-style causal attention on (B, H, S, D), GQA via n_kv_heads."""

    @staticmethod
    def forward(ctx, q, k, v, causal):
        m = require_ext()
        o, lse = m.attn_fwd(q, k, v, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        m = require_ext()
        dq, dk, dv = m.attn_bwd(do, q, k, v, o, lse, ctx.causal)
        return dq, dk, dv, None


def _attention_eager(q, k, v, causal):
    # GQA: expand kv heads
    Hq, Hkv = q.shape[1], k.shape[1]
    if Hkv != Hq:
        rep = Hq // Hkv
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    return F.scaled_dot_product_attention(q, k, v, is_causal=causal)


def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
              causal: bool = True) -> torch.Tensor:
    if q.is_cuda:
        return _Attention.apply(q, k, v, causal)
    return _attention_eager(q, k, v, causal)


class _AttentionLse(torch.autograd.Function):
    """Attention exposing the per-row logsumexp as a DIFFERENTIABLE
    second output — the building block of the context-parallel partial
    merge (scaffold/dist/cp.py).  d(lse) folds into the backward's
    delta term (delta' = rowsum(dO*O) - dlse) with zero extra kernels:
    dS gains  p * dlse_row  exactly (d lse / d s_k = p_k)."""

    @staticmethod
    def forward(ctx, q, k, v, causal):
        m = require_ext()
        o, lse = m.attn_fwd(q, k, v, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        return o, lse

    @staticmethod
    def backward(ctx, do, dlse):
        q, k, v, o, lse = ctx.saved_tensors
        m = require_ext()
        dq, dk, dv = m.attn_bwd(do, q, k, v, o, lse, ctx.causal,
                                dlse.contiguous().float())
        return dq, dk, dv, None


def _eager_attn_lse(q, k, v, causal):
    """fp32 reference attention returning (o, lse); fully
    autograd-differentiable (the CPU path of attention_lse and the
    ground truth of the CP tests)."""
    Hq, Hkv = q.shape[1], k.shape[1]
    if Hkv != Hq:
        rep = Hq // Hkv
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    scale = 1.0 / (q.shape[-1] ** 0.5)
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    if causal:
        S_, Sk = s.shape[-2], s.shape[-1]
        mask = torch.ones(S_, Sk, dtype=torch.bool, device=s.device).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    lse = torch.logsumexp(s, dim=-1)
    p = torch.exp(s - lse.unsqueeze(-1))
    o = (p @ v.float()).to(q.dtype)
    return o, lse


def attention_lse(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                  causal: bool = True):
    """(o, lse) with gradients through BOTH outputs."""
    if q.is_cuda:
        return _AttentionLse.apply(q, k, v, causal)
    return _eager_attn_lse(q, k, v, causal)


def _colsum(dy2: torch.Tensor) -> torch.Tensor:
    m = ext()
    if m is not None and dy2.is_cuda and dy2.dtype == torch.bfloat16:
        return m.colsum(dy2)
    return dy2.sum(0)


# ------------------------------------------------- K5 hand MFMA linear
def _tile_ok(x: torch.Tensor, w: torch.Tensor) -> bool:
    """The hand kernel covers the benched transformer-block shapes
    (SURVEY §2.3 K5): M%128, N%128, K%64.  Other shapes (e.g. the
    vocab-sized lm_head) take the hipBLASLt library-GEMM path, per
    SURVEY §2.2 ("hipBLASLt/rocBLAS for general GEMM")."""
    K = x.shape[-1]
    M = x.numel() // K
    return (x.dtype == torch.bfloat16 and w.dtype == torch.bfloat16
            and M % 128 == 0 and w.shape[0] % 128 == 0 and K % 64 == 0)


class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b):
        m = require_ext()
        x = x.contiguous()
        (y,) = m.linear_fwd(x, w, b, 0, None)
        ctx.save_for_backward(x, w)
        ctx.has_bias = b is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy2 = dy.contiguous().view(-1, dy.shape[-1])
        x2 = x.view(-1, x.shape[-1])
        dx = (dy2 @ w).view(x.shape)
        dw = dy2.t() @ x2
        db = _colsum(dy2) if ctx.has_bias else None
        return dx, dw, db


class _LinearGeluFn(torch.autograd.Function):
    """fc + bias + tanh-GELU in ONE kernel (the GPT-2 MLP up-projection).
    Saves the pre-activation for an exact GELU backward."""

    @staticmethod
    def forward(ctx, x, w, b):
        m = require_ext()
        x = x.contiguous()
        y, pre = m.linear_fwd(x, w, b, 1, None)
        ctx.save_for_backward(x, w, pre)
        ctx.has_bias = b is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, pre = ctx.saved_tensors
        dpre = torch.ops.aten.gelu_backward(dy, pre, approximate="tanh")
        dpre2 = dpre.contiguous().view(-1, dpre.shape[-1])
        x2 = x.view(-1, x.shape[-1])
        dx = (dpre2 @ w).view(x.shape)
        dw = dpre2.t() @ x2
        db = _colsum(dpre2) if ctx.has_bias else None
        return dx, dw, db


class _LinearResidualFn(torch.autograd.Function):
    """proj + bias + residual-add in ONE kernel (transformer residual
    branches) — removes a full elementwise add pass over HBM."""

    @staticmethod
    def forward(ctx, x, w, b, res):
        m = require_ext()
        x = x.contiguous()
        (y,) = m.linear_fwd(x, w, b, 2, res.contiguous())
        ctx.save_for_backward(x, w)
        ctx.has_bias = b is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy2 = dy.contiguous().view(-1, dy.shape[-1])
        x2 = x.view(-1, x.shape[-1])
        dx = (dy2 @ w).view(x.shape)
        dw = dy2.t() @ x2
        db = _colsum(dy2) if ctx.has_bias else None
        return dx, dw, db, dy


# Per-shape dispatch switch: measured on MI355X (gpurun_out/gemm_bench.log,
# 2026-09-13) the 2-phase 128^2 hand kernel reaches ~900 TF, below
# hipBLASLt at most benched shapes EXCEPT the small-N residual-fused
# proj (1.07x).  Until the 256^2 8-phase upgrade lands, only the
# shapes where the hand kernel measured >= parity route to it.
_HAND_GEMM_ALL = False  # flipped by benchmarks once the 8-phase core wins


class _LinearLib(torch.autograd.Function):
    """Library-GEMM (hipBLASLt) forward with an explicit backward:
    rocBLAS matmuls for dX/dW plus the hand colsum kernel for the bias
    gradient (torch's generic reduce measured ~1 TB/s there)."""

    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        ctx.has_bias = b is not None
        return F.linear(x, w, b)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy2 = dy.contiguous().view(-1, dy.shape[-1])
        x2 = x.contiguous().view(-1, x.shape[-1])
        dx = (dy2 @ w).view(x.shape)
        dw = dy2.t() @ x2
        db = None
        if ctx.has_bias:
            m = ext()
            if m is not None and dy2.dtype == torch.bfloat16:
                db = m.colsum(dy2)
            else:
                db = dy2.sum(0)
        return dx, dw, db


def linear(x: torch.Tensor, w: torch.Tensor, b: Optional[torch.Tensor] = None
           ) -> torch.Tensor:
    if x.is_cuda and _HAND_GEMM_ALL and _tile_ok(x, w):
        return _LinearFn.apply(x, w, b)
    if x.is_cuda and x.dtype == torch.bfloat16:
        return _LinearLib.apply(x, w, b)
    return F.linear(x, w, b)


def linear_gelu(x: torch.Tensor, w: torch.Tensor,
                b: Optional[torch.Tensor] = None) -> torch.Tensor:
    if x.is_cuda and _HAND_GEMM_ALL and _tile_ok(x, w):
        return _LinearGeluFn.apply(x, w, b)
    if x.is_cuda and x.dtype == torch.bfloat16:
        return F.gelu(_LinearLib.apply(x, w, b), approximate="tanh")
    return F.gelu(F.linear(x, w, b), approximate="tanh")


def linear_residual(x: torch.Tensor, w: torch.Tensor,
                    b: Optional[torch.Tensor], res: torch.Tensor) -> torch.Tensor:
    # measured >= hipBLASLt only for small N AND K (the square attn
    # proj); bigger shapes stay on the library GEMM + add
    if x.is_cuda and _tile_ok(x, w) and (
            _HAND_GEMM_ALL or (w.shape[0] <= 1024 and w.shape[1] <= 1024)):
        return _LinearResidualFn.apply(x, w, b, res)
    return F.linear(x, w, b) + res


# ----------------------------------------------- fused lm_head + CE
class _LmHeadCE(torch.autograd.Function):
    """Chunked fused final projection + softmax-cross-entropy.

    The full [T, V] logits matrix (1.65 GB for GPT-2, 4 GB for Llama at
    the bench shapes) never round-trips HBM: each row chunk's logits are
    produced (hipBLASLt), reduced to loss and dlogits (K3 kernels), and
    immediately consumed into dh and an fp32 dW accumulator while still
    cache-resident.  Since the logits are not kept, both grads are
    computed in forward (per-chunk bwd kernel scale R_c/T folds the mean
    over ALL T rows) and scaled by the incoming dloss in backward.
    """

    CHUNK = 2048

    @staticmethod
    def forward(ctx, x, w, targets):
        m = require_ext()
        h2 = x.contiguous().view(-1, x.shape[-1])
        ctx.x_shape = x.shape
        T = h2.shape[0]
        dh = torch.empty_like(h2)
        dw32 = torch.zeros_like(w, dtype=torch.float32)
        loss_sum = None
        for r0 in range(0, T, _LmHeadCE.CHUNK):
            hc = h2[r0:r0 + _LmHeadCE.CHUNK]
            tc = targets[r0:r0 + _LmHeadCE.CHUNK]
            logits = hc @ w.t()
            loss_c, lse = m.softmax_xent_fwd(logits, tc)
            s = loss_c.sum()
            loss_sum = s if loss_sum is None else loss_sum + s
            dlogits = m.softmax_xent_bwd(logits, tc, lse, hc.shape[0] / T)
            dh[r0:r0 + _LmHeadCE.CHUNK] = dlogits @ w
            dw32 += (dlogits.t() @ hc).float()
        ctx.saved_dh = dh
        ctx.saved_dw32 = dw32
        ctx.w_dtype = w.dtype
        return loss_sum / T

    @staticmethod
    def backward(ctx, dloss):
        dh = (ctx.saved_dh * dloss).view(ctx.x_shape)
        dw = (ctx.saved_dw32 * dloss.float()).to(ctx.w_dtype)
        ctx.saved_dh = None
        ctx.saved_dw32 = None
        return dh, dw, None


import os as _os_env

# Measured A/B (2x25-step runs each, GPT-2 bs16): unfused 444.4
# samples/s vs fused-chunked 406.4 — the fp32 dW accumulator traffic
# (2.5 GB/step) and the smaller per-chunk GEMMs cost more than the
# logits HBM round-trip saves on THIS vocab/batch.  Unfused is the
# default; the chunked path stays for memory-constrained regimes
# (SCAFFOLD_FUSED_CE=1) where the [T, V] logits would not fit.
_FUSED_CE = _os_env.environ.get("SCAFFOLD_FUSED_CE", "0") == "1"


def lm_head_ce(x: torch.Tensor, w: torch.Tensor, targets: torch.Tensor
               ) -> torch.Tensor:
    """mean CE of (x @ w^T) against targets; x (..., D), targets (...)."""
    t = targets.contiguous().view(-1)
    if x.is_cuda and _FUSED_CE:
        return _LmHeadCE.apply(x, w, t)
    if x.is_cuda:
        # unfused: full logits through blaslt + K3 CE kernels
        return softmax_cross_entropy(linear(x, w), t)
    return F.cross_entropy(
        x.contiguous().view(-1, x.shape[-1]).float() @ w.float().t(), t)


# ----------------------------------------------------------------- SwiGLU
class _SwiGLU(torch.autograd.Function):
    """Fused silu(gate)*up (Llama MLP) — one HBM pass each direction."""

    @staticmethod
    def forward(ctx, g, u):
        m = require_ext()
        g = g.contiguous()
        u = u.contiguous()
        ctx.save_for_backward(g, u)
        return m.swiglu_fwd(g, u)

    @staticmethod
    def backward(ctx, dh):
        g, u = ctx.saved_tensors
        m = require_ext()
        dg, du = m.swiglu_bwd(dh.contiguous(), g, u)
        return dg, du


def swiglu(g: torch.Tensor, u: torch.Tensor) -> torch.Tensor:
    if g.is_cuda and g.dtype == torch.bfloat16 and g.numel() % 8 == 0:
        return _SwiGLU.apply(g, u)
    return F.silu(g) * u


class _SwiGLUPacked(torch.autograd.Function):
    """SwiGLU over ONE packed [.., 2F] gate|up buffer (the fused
    gate+up GEMM output) — bwd emits one packed dgu for a single
    weight-grad GEMM."""

    @staticmethod
    def forward(ctx, gu):
        m = require_ext()
        gu = gu.contiguous()
        ctx.save_for_backward(gu)
        return m.swiglu_packed_fwd(gu)

    @staticmethod
    def backward(ctx, dh):
        (gu,) = ctx.saved_tensors
        m = require_ext()
        return m.swiglu_packed_bwd(dh.contiguous(), gu)


def swiglu_packed(gu: torch.Tensor) -> torch.Tensor:
    """h = silu(gu[..., :F]) * gu[..., F:] where F = gu.shape[-1]//2."""
    if gu.is_cuda and gu.dtype == torch.bfloat16 and gu.shape[-1] % 16 == 0:
        return _SwiGLUPacked.apply(gu)
    F_ = gu.shape[-1] // 2
    return F.silu(gu[..., :F_]) * gu[..., F_:]


# ------------------------------------------------------------- fused AdamW
def fused_adamw_step(
    params_fp32: list,
    grads: list,
    exp_avgs: list,
    exp_avg_sqs: list,
    params_bf16: Optional[list],
    *,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    step: int,
    grad_scale: float = 1.0,
) -> None:
    """One decoupled-AdamW step over many tensors in one kernel launch.

    fp32 master params; grads may be bf16 or fp32; optionally also writes
    the bf16 working copy (params_bf16) in the same pass (K8 folded into
    K1 per SURVEY §2.3).
    """
    if params_fp32 and params_fp32[0].is_cuda:
        m = require_ext()
        m.fused_adamw(
            params_fp32, grads, exp_avgs, exp_avg_sqs,
            params_bf16 if params_bf16 is not None else [],
            lr, beta1, beta2, eps, weight_decay, step, grad_scale,
        )
        return
    # CPU eager reference (used by tests as ground truth)
    bc1 = 1.0 - beta1 ** step
    bc2 = 1.0 - beta2 ** step
    for i, p in enumerate(params_fp32):
        g = grads[i].float() * grad_scale
        ea, eas = exp_avgs[i], exp_avg_sqs[i]
        p.mul_(1.0 - lr * weight_decay)
        ea.mul_(beta1).add_(g, alpha=1 - beta1)
        eas.mul_(beta2).addcmul_(g, g, value=1 - beta2)
        denom = (eas / bc2).sqrt_().add_(eps)
        p.addcdiv_(ea / bc1, denom, value=-lr)
        if params_bf16 is not None and params_bf16:
            params_bf16[i].copy_(p.to(params_bf16[i].dtype))

# ------------------------------------------------- fused BatchNorm(+ReLU)
class _BatchNormAct(torch.autograd.Function):
    """Training-mode bf16 NHWC BatchNorm with fused ReLU and optional
    fused residual add (csrc/kernels/batchnorm.hip).  Replaces MIOpen's
    fp32-only BN + the bf16<->fp32 cast copies + separate add/relu
    passes that together were ~57%% of the r1 ResNet step
    (profiles/resnet_kstats_r2.txt)."""

    @staticmethod
    def forward(ctx, x, res, weight, bias, running_mean, running_var,
                momentum, eps, relu):
        m = require_ext()
        y, mean, rstd = m.bn_fwd(x, res, weight, bias, running_mean,
                                 running_var, momentum, eps, relu)
        ctx.save_for_backward(x, weight, bias, mean, rstd,
                              res if res is not None else x.new_empty(0))
        ctx.relu = relu
        ctx.has_res = res is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, bias, mean, rstd, res = ctx.saved_tensors
        m = require_ext()
        out = m.bn_bwd(dy, x, res if ctx.has_res else None, weight, bias,
                       mean, rstd, ctx.relu)
        if ctx.has_res:
            dx, dres, dw, db = out
        else:
            (dx, dw, db), dres = out, None
        return dx, dres, dw, db, None, None, None, None, None


def _bn_act_eager(x, res, weight, bias, running_mean, running_var,
                  momentum, eps, relu, training):
    """CPU/eval reference: fp32 BN + add + relu."""
    y = F.batch_norm(x.float(), running_mean, running_var, weight, bias,
                     training, momentum, eps)
    if res is not None:
        y = y + res.float()
    if relu:
        y = F.relu(y)
    return y.to(x.dtype)


def batch_norm_act(x: torch.Tensor, res: Optional[torch.Tensor],
                   weight: torch.Tensor, bias: torch.Tensor,
                   running_mean: torch.Tensor, running_var: torch.Tensor,
                   momentum: float, eps: float, relu: bool,
                   training: bool) -> torch.Tensor:
    if x.is_cuda and training:
        return _BatchNormAct.apply(
            x.contiguous(memory_format=torch.channels_last),
            None if res is None else res.contiguous(memory_format=torch.channels_last),
            weight, bias, running_mean, running_var, momentum, eps, relu)
    # CPU path and eval mode (running-stat normalize): eager reference
    return _bn_act_eager(x, res, weight, bias, running_mean, running_var,
                         momentum, eps, relu, training)

