"""Synthetic-data determinism tier (SURVEY §4 unit)."""

import torch

from scaffold.config import load_config
from scaffold.data.synthetic import build_data


def _cfg(kind, extra=()):
    base = {
        "synthetic_tokens": ["model.name=gpt2", "model.vocab_size=128",
                             "data.seq_len=16", "data.batch_size=4"],
        "synthetic_vector": ["model.name=mlp", "model.in_dim=8",
                             "model.out_dim=4", "data.vector_dim=8",
                             "data.batch_size=4"],
    }[kind]
    return load_config(None, [f"data.kind={kind}", "device=cpu", *base, *extra])


def test_same_seed_same_stream():
    cfg = _cfg("synthetic_tokens")
    a = build_data(cfg, 0, torch.device("cpu"), torch.float32)
    b = build_data(cfg, 0, torch.device("cpu"), torch.float32)
    for _ in range(3):
        xa, ya = a.next_batch()
        xb, yb = b.next_batch()
        assert torch.equal(xa, xb) and torch.equal(ya, yb)


def test_ranks_see_distinct_data():
    cfg = _cfg("synthetic_tokens")
    a = build_data(cfg, 0, torch.device("cpu"), torch.float32)
    b = build_data(cfg, 1, torch.device("cpu"), torch.float32)
    xa, _ = a.next_batch()
    xb, _ = b.next_batch()
    assert not torch.equal(xa, xb)


def test_state_roundtrip_resumes_stream():
    cfg = _cfg("synthetic_vector")
    a = build_data(cfg, 0, torch.device("cpu"), torch.float32)
    a.next_batch()
    sd = a.state_dict()
    x1, y1 = a.next_batch()
    b = build_data(cfg, 0, torch.device("cpu"), torch.float32)
    b.load_state_dict(sd)
    x2, y2 = b.next_batch()
    assert torch.equal(x1, x2) and torch.equal(y1, y2)


def test_active_vocab_restricts_tokens():
    cfg = _cfg("synthetic_tokens", ["data.active_vocab=7"])
    d = build_data(cfg, 0, torch.device("cpu"), torch.float32)
    x, y = d.next_batch()
    assert int(x.max()) < 7 and int(y.max()) < 7
