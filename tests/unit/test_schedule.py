"""LR schedule + grad-clip unit tier."""

import torch

from scaffold.config import load_config
from scaffold.schedule import lr_at
from scaffold.trainer import Trainer


def test_warmup_and_cosine_shape():
    cfg = load_config(None, [
        "optim.lr=0.1", "optim.schedule=cosine", "optim.warmup_steps=10",
        "optim.min_lr_ratio=0.1", "steps=110",
    ])
    lrs = [lr_at(cfg, s, cfg.steps) for s in range(110)]
    assert abs(lrs[0] - 0.01) < 1e-9          # warmup start lr/10
    assert abs(lrs[9] - 0.1) < 1e-9           # warmup end
    assert abs(lrs[10] - 0.1) < 1e-6          # cosine start
    assert lrs[60] < lrs[10]                  # decaying
    assert abs(lrs[109] - 0.01) > -1          # floor-ish
    assert min(lrs) >= 0.01 - 1e-6            # never below floor


def test_linear_and_const():
    cfg = load_config(None, ["optim.lr=1.0", "optim.schedule=linear",
                             "optim.min_lr_ratio=0.5", "steps=100"])
    assert abs(lr_at(cfg, 50, 100) - 0.75) < 1e-6
    cfg2 = load_config(None, ["optim.lr=0.3", "steps=10"])
    assert lr_at(cfg2, 7, 10) == 0.3


def test_grad_clip_limits_update():
    def run(clip):
        cfg = load_config(None, [
            "model.name=mlp", "model.in_dim=16", "model.hidden_dim=32",
            "model.out_dim=4", "data.kind=synthetic_vector",
            "data.batch_size=8", "data.vector_dim=16", "dtype=fp32",
            "device=cpu", "steps=1", "optim.lr=1.0", "log.interval=100",
            f"optim.grad_clip={clip}",
        ])
        t = Trainer(cfg)
        p0 = torch.cat([p.detach().flatten().clone() for p in t.raw_model.parameters()])
        it = iter(t.data)
        t._train_step(it, 1)
        p1 = torch.cat([p.detach().flatten() for p in t.raw_model.parameters()])
        return (p1 - p0).abs().max().item()

    big = run(0.0)
    small = run(1e-4)
    assert small < big, (small, big)
