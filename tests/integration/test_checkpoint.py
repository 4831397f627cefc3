"""Checkpoint round-trip (C3): save -> resume -> bit-exact continuation."""

import copy

import torch

from scaffold.ckpt.layout import latest_checkpoint, read_manifest
from scaffold.trainer import Trainer


def _train_n(t, n):
    it = iter(t.data)
    last = 0.0
    for _ in range(n):
        t.ddp.zero_grad()
        x, y = next(it)
        loss = t.model(x, y)
        loss.backward()
        t.ddp.finish()
        t.optimizer.step(grad_scale=t.ddp.grad_scale)
        t.step += 1
        last = loss.item()
    return last


def test_save_resume_bitexact(cpu_cfg, tmp_path):
    cfg = copy.deepcopy(cpu_cfg)
    cfg.ckpt.dir = str(tmp_path)
    t = Trainer(cfg)
    _train_n(t, 3)
    ck = t.save_checkpoint()

    m = read_manifest(ck)
    assert m["step"] == 3
    assert m["world_size"] == 1
    assert latest_checkpoint(str(tmp_path)) == ck

    # continue original 2 more steps
    _train_n(t, 2)
    ref = [p.detach().clone() for p in t.raw_model.parameters()]

    # fresh trainer resumes from ckpt, runs the same 2 steps
    cfg2 = copy.deepcopy(cfg)
    cfg2.ckpt.resume = ck
    t2 = Trainer(cfg2)
    assert t2.step == 3
    assert t2.optimizer.step_count == t.optimizer.step_count - 2
    _train_n(t2, 2)
    for a, b in zip(ref, t2.raw_model.parameters()):
        assert torch.equal(a, b.detach()), "resume continuation not bit-exact"


def test_manifest_missing_is_incomplete(tmp_path):
    import pytest

    from scaffold.ckpt.layout import read_manifest

    d = tmp_path / "step_00000001"
    d.mkdir()
    with pytest.raises(FileNotFoundError):
        read_manifest(str(d))
    assert latest_checkpoint(str(tmp_path)) == ""


def test_reshard_by_name(cpu_cfg, tmp_path):
    """Load a world_size=1 checkpoint into a fresh world_size=1 trainer
    with different RNG position (by-name reshard path)."""
    cfg = copy.deepcopy(cpu_cfg)
    cfg.ckpt.dir = str(tmp_path)
    t = Trainer(cfg)
    _train_n(t, 2)
    ck = t.save_checkpoint()

    cfg2 = copy.deepcopy(cfg)
    cfg2.seed = 999  # different init; load must overwrite everything
    cfg2.ckpt.resume = ck
    t2 = Trainer(cfg2)
    for a, b in zip(t.raw_model.parameters(), t2.raw_model.parameters()):
        assert torch.equal(a.detach(), b.detach())
    for a, b in zip(t.optimizer.exp_avg, t2.optimizer.exp_avg):
        assert torch.equal(a, b)
