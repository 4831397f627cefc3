"""LoRA adapters (scaffold/lora.py): identity at init, adapters-only
training, merge equivalence, portable adapter state, and checkpoint
alignment with frozen params present."""

import pytest
import torch

from scaffold.config import load_config
from scaffold.lora import (apply_lora, load_lora_state_dict,
                           lora_state_dict, merge_lora)
from scaffold.models import build_model


def _cfg(lora=0, **kw):
    over = [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=2",
        "model.d_model=32", "model.vocab_size=64", "model.max_seq_len=16",
        f"model.lora_r={lora}", "data.kind=synthetic_tokens",
        "data.batch_size=4", "data.seq_len=16", "data.active_vocab=8",
        "device=cpu", "dtype=fp32", "log.interval=1000", "optim.lr=0.02",
    ]
    over += [f"{k}={v}" for k, v in kw.items()]
    return load_config(None, over)


def test_identity_at_init_and_trainable_set():
    torch.manual_seed(3)
    m = build_model(_cfg())
    tok = torch.randint(0, 64, (2, 16))
    ref = m(tok, tok).detach()
    n = apply_lora(m, r=4, alpha=8)
    assert n > 0
    # B starts at zero -> bitwise identity
    assert torch.equal(m(tok, tok).detach(), ref)
    trainable = [p for p in m.parameters() if p.requires_grad]
    total = list(m.parameters())
    assert 0 < len(trainable) < len(total)
    assert all(p.numel() <= 4 * (32 + 128) for p in trainable)  # rank-sized


def test_training_touches_only_adapters_and_learns():
    from scaffold.trainer import Trainer

    # adapters only (embeddings + lm_head frozen) on a tiny model:
    # capacity is low, so expect a modest but definite decline
    t = Trainer(_cfg(lora=8, steps=60, **{"optim.lr": 0.05,
                                          "model.lora_alpha": 32}))
    frozen_before = [p.detach().clone()
                     for p in t.raw_model.parameters() if not p.requires_grad]
    it = iter(t.data)
    losses = [t._train_step(it, 1) for _ in range(60)]
    assert min(losses[-5:]) < losses[0] - 0.08, (losses[0], losses[-5:])
    frozen_after = [p.detach()
                    for p in t.raw_model.parameters() if not p.requires_grad]
    for a, b in zip(frozen_before, frozen_after):
        assert torch.equal(a, b)


def test_merge_matches_adapter_forward():
    torch.manual_seed(5)
    m = build_model(_cfg())
    apply_lora(m, r=4, alpha=8)
    # perturb adapters so the merge is non-trivial
    with torch.no_grad():
        for n_, p in m.named_parameters():
            if p.requires_grad:
                p.add_(torch.randn_like(p) * 0.05)
    tok = torch.randint(0, 64, (2, 16))
    ref = m(tok, tok).detach()
    sd = lora_state_dict(m)
    assert sd and all(v.numel() < 4 * 200 for v in sd.values())
    merge_lora(m)
    assert not any("parametrizations" in k for k in m.state_dict())
    assert torch.allclose(m(tok, tok).detach(), ref, rtol=1e-5, atol=1e-6)

    # adapters re-load into a fresh copy and reproduce the same model
    torch.manual_seed(5)
    m2 = build_model(_cfg())
    apply_lora(m2, r=4, alpha=8)
    load_lora_state_dict(m2, sd)
    assert torch.allclose(m2(tok, tok).detach(), ref, rtol=1e-5, atol=1e-6)


def test_lora_checkpoint_roundtrip(tmp_path):
    """Frozen params + adapters through the sharded checkpoint engine:
    the optimizer's name identities keep opt state aligned."""
    import copy

    from scaffold.trainer import Trainer

    cfg = _cfg(lora=4, steps=3)
    cfg.ckpt.dir = str(tmp_path)
    t = Trainer(cfg)
    it = iter(t.data)
    for _ in range(3):
        t._train_step(it, 1)
        t.step += 1
    ck = t.save_checkpoint()

    cfg2 = copy.deepcopy(cfg)
    cfg2.ckpt.resume = ck
    cfg2.seed = 9
    t2 = Trainer(cfg2)
    assert t2.step == 3
    for (n, a), (_, b) in zip(t.raw_model.named_parameters(),
                              t2.raw_model.named_parameters()):
        assert torch.equal(a.detach(), b.detach()), n
    for ma, mb in zip(t.optimizer.master, t2.optimizer.master):
        assert torch.equal(ma, mb)


def test_init_weights_then_lora_finetune(tmp_path):
    """The fine-tune flow end to end: pretrained weights file ->
    ckpt.init_weights -> LoRA training starts FROM those weights."""
    from scaffold.convert import save_model
    from scaffold.trainer import Trainer

    torch.manual_seed(21)
    pre = build_model(_cfg())
    p = str(tmp_path / "pre.safetensors")
    save_model(pre, p)

    cfg = _cfg(lora=4, steps=1, **{"ckpt.init_weights": p, "seed": 5})
    t = Trainer(cfg)
    # adapters start at identity, so forward equality proves the base
    # weights came from the pretrained file, not a fresh seed-5 init
    tok = torch.randint(0, 64, (2, 16))
    assert torch.equal(t.raw_model(tok, tok).detach(),
                       pre(tok, tok).detach())


def test_init_weights_quantized(tmp_path):
    from scaffold.quant import save_quantized
    from scaffold.trainer import Trainer

    torch.manual_seed(23)
    pre = build_model(_cfg())
    p = str(tmp_path / "pre_q.safetensors")
    save_quantized(pre, p)
    t = Trainer(_cfg(**{"ckpt.init_weights": p, "seed": 6}))
    tok = torch.randint(0, 64, (2, 16))
    a = t.raw_model(tok, tok).detach()
    b = pre(tok, tok).detach()
    assert torch.allclose(a, b, rtol=0.05, atol=0.05)


@pytest.mark.gpu
def test_lora_gpu_identity_and_step():
    """LoRA on MI355X: adapters at init are a bitwise no-op through the
    HIP op path, and a train step moves ONLY adapter params."""
    from scaffold.trainer import Trainer

    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=256", "model.vocab_size=512", "model.max_seq_len=128",
        "data.kind=synthetic_tokens", "data.batch_size=4", "data.seq_len=128",
        "data.active_vocab=16", "dtype=bf16", "log.interval=1000",
        "optim.lr=0.01",
    ])
    torch.manual_seed(7)
    base = build_model(cfg).to("cuda", torch.bfloat16).eval()
    tok = torch.randint(0, 512, (2, 128), device="cuda")
    ref = base(tok, tok).detach()
    apply_lora(base, r=8, alpha=16)
    assert torch.equal(base(tok, tok).detach(), ref)

    cfg.model.lora_r = 8
    t = Trainer(cfg)
    frozen = [p.detach().clone() for p in t.raw_model.parameters()
              if not p.requires_grad]
    it = iter(t.data)
    losses = [t._train_step(it, 1) for _ in range(5)]
    torch.cuda.synchronize()
    assert all(torch.isfinite(torch.tensor(losses))), losses
    after = [p.detach() for p in t.raw_model.parameters()
             if not p.requires_grad]
    for a, b in zip(frozen, after):
        assert torch.equal(a, b)


def test_lora_on_moe_model_freezes_experts():
    """LoRA over a MoE model: experts (raw Parameters, not nn.Linear)
    freeze with the backbone; adapters train; loss still decreases."""
    from scaffold.trainer import Trainer

    cfg = _cfg(lora=8, steps=40, **{"model.n_experts": 4,
                                    "optim.lr": 0.05,
                                    "model.lora_alpha": 32})
    t = Trainer(cfg)
    moe = t.raw_model.blocks[0].moe
    assert not any(p.requires_grad
                   for p in list(moe.w_up) + list(moe.w_down))
    frozen = [p.detach().clone()
              for p in list(moe.w_up) + list(moe.w_down)]
    it = iter(t.data)
    losses = [t._train_step(it, 1) for _ in range(40)]
    assert min(losses[-5:]) < losses[0] - 0.05, (losses[0], losses[-5:])
    for a, b in zip(frozen, list(moe.w_up) + list(moe.w_down)):
        assert torch.equal(a, b.detach())
