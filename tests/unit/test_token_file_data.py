"""TokenFileData (data.kind=token_file): memory-mapped corpus windows
with per-rank determinism, resume exactness, and end-to-end training."""

import numpy as np
import pytest
import torch

from scaffold.config import load_config
from scaffold.data.synthetic import TokenFileData, build_data


def _corpus(tmp_path, n=4096, vocab=128):
    rng = np.random.default_rng(0)
    arr = rng.integers(0, vocab, size=n, dtype=np.uint16)
    p = tmp_path / "corpus.bin"
    arr.tofile(p)
    return str(p), arr


def _cfg(path, **kw):
    over = [
        "model.name=gpt2", "model.n_layers=1", "model.n_heads=2",
        "model.d_model=32", "model.vocab_size=128", "model.max_seq_len=16",
        "data.kind=token_file", f"data.path={path}",
        "data.batch_size=4", "data.seq_len=16", "device=cpu", "dtype=fp32",
    ]
    over += [f"{k}={v}" for k, v in kw.items()]
    return load_config(None, over)


def test_windows_come_from_corpus(tmp_path):
    path, arr = _corpus(tmp_path)
    d = build_data(_cfg(path), 0, torch.device("cpu"), torch.float32)
    assert isinstance(d, TokenFileData)
    x, y = d.next_batch()
    assert x.shape == (4, 16) and y.shape == (4, 16)
    # y is x shifted by one within a contiguous corpus window
    assert torch.equal(x[:, 1:], y[:, :-1])
    flat = arr.astype(np.int64)
    for row_x, row_y in zip(x, y):
        w = torch.cat([row_x, row_y[-1:]]).numpy()
        # the window exists verbatim somewhere in the corpus
        found = any(np.array_equal(flat[s:s + 17], w)
                    for s in range(len(flat) - 17))
        assert found


def test_ranks_see_distinct_streams(tmp_path):
    path, _ = _corpus(tmp_path)
    cfg = _cfg(path)
    d0 = TokenFileData(cfg, 0, torch.device("cpu"), torch.float32)
    d1 = TokenFileData(cfg, 1, torch.device("cpu"), torch.float32)
    x0, _ = d0.next_batch()
    x1, _ = d1.next_batch()
    assert not torch.equal(x0, x1)


def test_resume_exactness(tmp_path):
    path, _ = _corpus(tmp_path)
    cfg = _cfg(path)
    d = TokenFileData(cfg, 0, torch.device("cpu"), torch.float32)
    d.next_batch()
    snap = d.state_dict()
    a = [d.next_batch()[0] for _ in range(3)]
    d.load_state_dict(snap)
    b = [d.next_batch()[0] for _ in range(3)]
    for t0, t1 in zip(a, b):
        assert torch.equal(t0, t1)


def test_trains_and_checkpoints(tmp_path):
    from scaffold.trainer import Trainer

    path, _ = _corpus(tmp_path)
    cfg = _cfg(path, **{"log.interval": 1000, "steps": 3,
                        "ckpt.dir": str(tmp_path / "ck"),
                        "ckpt.save_final": "true"})
    t = Trainer(cfg)
    t.fit()
    assert t.step == 3
    # resume from the final checkpoint restores the data cursor
    from scaffold.ckpt import layout
    last = layout.latest_checkpoint(str(tmp_path / "ck"))
    cfg2 = _cfg(path, **{"log.interval": 1000, "steps": 3,
                         "ckpt.resume": last})
    t2 = Trainer(cfg2)
    assert t2.step == 3
    xa, _ = t.data.next_batch()
    xb, _ = t2.data.next_batch()
    assert torch.equal(xa, xb)


def test_rejects_missing_or_short_file(tmp_path):
    with pytest.raises(Exception):
        build_data(_cfg(str(tmp_path / "nope.bin")), 0,
                   torch.device("cpu"), torch.float32)
    p = tmp_path / "short.bin"
    np.zeros(8, dtype=np.uint16).tofile(p)
    with pytest.raises(ValueError, match="need >"):
        build_data(_cfg(str(p)), 0, torch.device("cpu"), torch.float32)


def test_uint32_corpus_for_large_vocab(tmp_path):
    """vocab > 65536 switches the memmap dtype to uint32."""
    rng = np.random.default_rng(1)
    arr = rng.integers(0, 100_000, size=2048, dtype=np.uint32)
    p = tmp_path / "big.bin"
    arr.tofile(p)
    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=1", "model.n_heads=2",
        "model.d_model=32", "model.vocab_size=128256",
        "model.max_seq_len=16",
        "data.kind=token_file", f"data.path={p}",
        "data.batch_size=2", "data.seq_len=16", "device=cpu", "dtype=fp32",
    ])
    d = build_data(cfg, 0, torch.device("cpu"), torch.float32)
    x, y = d.next_batch()
    assert x.dtype == torch.int64
    assert int(x.max()) < 100_000
    assert torch.equal(x[:, 1:], y[:, :-1])
