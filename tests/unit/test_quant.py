"""Int8 weight-only export (scaffold/quant.py): size, error bounds,
roundtrip into a model, and tied-tensor handling."""

import os

import pytest
import torch

from scaffold.config import load_config
from scaffold.models import build_model
from scaffold.quant import (dequantize_tensor, load_quantized,
                            quantize_tensor, save_quantized)


def _model():
    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=64", "model.vocab_size=128", "model.max_seq_len=32",
    ])
    torch.manual_seed(71)
    return build_model(cfg).eval()


def test_per_channel_error_bound():
    torch.manual_seed(0)
    w = torch.randn(64, 128) * torch.logspace(-2, 1, 64).unsqueeze(1)
    q, s = quantize_tensor(w)
    back = dequantize_tensor(q, s)
    # each row's max abs error is at most half a step of ITS OWN scale
    err = (back - w).abs().amax(dim=1)
    assert torch.all(err <= s.squeeze(1) * 0.5 + 1e-8)


def test_roundtrip_close_and_smaller(tmp_path):
    from scaffold.convert import save_model

    m = _model()
    pq = str(tmp_path / "q.safetensors")
    pf = str(tmp_path / "f.safetensors")
    save_quantized(m, pq)
    save_model(m, pf)
    # int8 + fp32 scales vs fp32 weights: ~4x smaller on the 2-D bulk
    assert os.path.getsize(pq) < 0.45 * os.path.getsize(pf)

    m2 = _model()
    with torch.no_grad():
        for p in m2.parameters():
            p.zero_()
    load_quantized(m2, pq)
    # weights close per-channel; logits close
    for (n, a), (_, b) in zip(m.state_dict().items(),
                              m2.state_dict().items()):
        assert torch.allclose(a, b, atol=0.05, rtol=0.05), n
    tok = torch.randint(0, 128, (2, 16))
    la = m(tok, tok)
    lb = m2(tok, tok)
    assert torch.allclose(la, lb, rtol=0.05, atol=0.05), (la, lb)
    # tie preserved structurally
    assert m2.lm_head.weight.data_ptr() == m2.wte.weight.data_ptr()


@pytest.mark.gpu
def test_quantized_load_gpu(tmp_path):
    """int8 export loads into a bf16 CUDA model; logits stay close."""
    m = _model()
    p = str(tmp_path / "q.safetensors")
    save_quantized(m, p)
    mg = _model().to("cuda", torch.bfloat16)
    load_quantized(mg, p)
    tok = torch.randint(0, 128, (2, 16), device="cuda")
    la = mg(tok, tok).float().cpu()
    lb = m(tok, tok).float()
    torch.cuda.synchronize()
    assert torch.allclose(la, lb, rtol=0.1, atol=0.1), (la, lb)
