"""Property-based unit tier (hypothesis — SURVEY §4)."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from scaffold.config import TrainConfig, config_from_dict, config_to_dict, load_config
from scaffold.dist.ddp import partition_buckets
from scaffold.schedule import lr_at


class _P:
    """Stand-in parameter: numel/element_size/dtype like nn.Parameter."""

    def __init__(self, numel, dtype):
        self._n = numel
        self.dtype = dtype

    def numel(self):
        return self._n

    def element_size(self):
        return 2 if self.dtype == torch.bfloat16 else 4


@settings(max_examples=200, deadline=None)
@given(
    sizes=st.lists(st.integers(1, 5_000_000), min_size=1, max_size=64),
    dtypes=st.lists(st.sampled_from([torch.bfloat16, torch.float32]),
                    min_size=1, max_size=64),
    cap=st.floats(0.01, 64),
    first=st.floats(0.001, 8),
)
def test_partition_covers_every_param_once(sizes, dtypes, cap, first):
    params = [_P(n, dtypes[i % len(dtypes)]) for i, n in enumerate(sizes)]
    buckets = partition_buckets(params, cap, first)
    flat = [p for b in buckets for p in b]
    assert sorted(map(id, flat)) == sorted(map(id, params))  # exact cover
    for b in buckets:
        assert len({p.dtype for p in b}) == 1  # dtype-homogeneous
        # no bucket except possibly singletons exceeds the cap
        nbytes = sum(p.numel() * p.element_size() for p in b)
        if len(b) > 1:
            assert nbytes <= max(cap, first) * (1 << 20) + max(
                p.numel() * p.element_size() for p in b)


@settings(max_examples=100, deadline=None)
@given(
    lr=st.floats(1e-6, 1.0),
    warmup=st.integers(0, 50),
    total=st.integers(51, 500),
    ratio=st.floats(0.0, 1.0),
    sched=st.sampled_from(["const", "cosine", "linear"]),
)
def test_schedule_bounds(lr, warmup, total, ratio, sched):
    cfg = load_config(None, [
        f"optim.lr={lr}", f"optim.warmup_steps={warmup}",
        f"optim.schedule={sched}", f"optim.min_lr_ratio={ratio}",
        f"steps={total}",
    ])
    floor = 0.0 if sched == "const" else lr * ratio
    for s in range(0, total, max(1, total // 37)):
        v = lr_at(cfg, s, total)
        assert 0.0 <= v <= lr * (1 + 1e-9)
        if s >= warmup:
            assert v >= min(floor, lr) - 1e-12


@settings(max_examples=100, deadline=None)
@given(
    n_layers=st.integers(1, 48),
    bsz=st.integers(1, 512),
    steps=st.integers(1, 10000),
    clip=st.floats(0, 10),
)
def test_config_roundtrip(n_layers, bsz, steps, clip):
    cfg = config_from_dict({
        "model": {"name": "gpt2", "n_layers": n_layers},
        "data": {"batch_size": bsz},
        "optim": {"grad_clip": clip},
        "steps": steps,
    })
    d = config_to_dict(cfg)
    cfg2 = config_from_dict({k: v for k, v in d.items()
                             if k in ("model", "data", "optim", "dist",
                                      "ckpt", "log") or not isinstance(v, dict)})
    assert cfg2.model.n_layers == n_layers
    assert cfg2.data.batch_size == bsz
    assert cfg2.steps == steps
    assert abs(cfg2.optim.grad_clip - clip) < 1e-12


@settings(max_examples=100, deadline=None)
@given(
    b=st.integers(1, 4),
    v=st.integers(4, 200),
    top_k=st.integers(0, 220),
    top_p=st.floats(0.0, 1.0),
    temperature=st.floats(0.0, 3.0),
    seed=st.integers(0, 2**16),
)
def test_sample_respects_truncations(b, v, top_k, top_p, temperature, seed):
    """Generator._sample: ids always valid; temperature 0 is argmax;
    with top_k the draw is inside the k best logits."""
    from scaffold.generate import Generator

    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(b, v, generator=g)
    out = Generator._sample(logits, temperature, top_k, top_p,
                            torch.Generator().manual_seed(seed + 1))
    assert out.shape == (b, 1)
    assert int(out.min()) >= 0 and int(out.max()) < v
    if temperature <= 1e-5:
        assert torch.equal(out[:, 0], logits.argmax(dim=-1))
    elif 0 < top_k < v:
        kth = logits.topk(top_k, dim=-1).values[:, -1]
        chosen = logits.gather(1, out)[:, 0]
        assert bool((chosen >= kth - 1e-6).all())


@settings(max_examples=100, deadline=None)
@given(
    names_sizes=st.lists(
        st.tuples(st.integers(0, 10**9), st.integers(1, 10**8)),
        min_size=1, max_size=64, unique_by=lambda t: t[0]),
    world=st.integers(1, 8),
)
def test_shard_partition_covers_once(names_sizes, world):
    """ckpt layout: every tensor lands on exactly one shard."""
    from scaffold.ckpt.layout import partition_by_size

    named = [(f"t{n}", sz) for n, sz in names_sizes]
    assign = partition_by_size(named, world)
    assert sorted(assign.keys()) == sorted(n for n, _ in named)
    assert all(0 <= r < world for r in assign.values())


@settings(max_examples=100, deadline=None)
@given(vals=st.lists(st.floats(0, 1e6), min_size=0, max_size=50),
       q=st.floats(0.0, 1.0))
def test_percentile_bounds(vals, q):
    from scaffold.serve import _percentile

    sv = sorted(vals)
    p = _percentile(sv, q)
    if not sv:
        assert p == 0.0
    else:
        assert sv[0] <= p <= sv[-1]


@settings(max_examples=60, deadline=None)
@given(
    n_heads=st.integers(1, 16),
    hd_half=st.integers(1, 16),
    in_dim=st.integers(1, 32),
    seed=st.integers(0, 2**16),
)
def test_rope_permute_roundtrip(n_heads, hd_half, in_dim, seed):
    """HF<->interleaved RoPE row permutations are exact inverses."""
    from scaffold.convert import _permute_rope, _unpermute_rope

    g = torch.Generator().manual_seed(seed)
    w = torch.randn(n_heads * hd_half * 2, in_dim, generator=g)
    assert torch.equal(_permute_rope(_unpermute_rope(w, n_heads), n_heads), w)
    assert torch.equal(_unpermute_rope(_permute_rope(w, n_heads), n_heads), w)


@settings(max_examples=40, deadline=None)
@given(
    n=st.integers(40, 2000),
    seq=st.integers(1, 32),
    bsz=st.integers(1, 6),
    vocab=st.integers(2, 60000),
    seed=st.integers(0, 2**16),
)
def test_token_file_windows_always_valid(tmp_path_factory, n, seq, bsz,
                                         vocab, seed):
    """token_file batches: right shapes, ids from the corpus range,
    y == x shifted, for arbitrary corpus/seq/batch geometry."""
    import numpy as np

    from scaffold.data.synthetic import TokenFileData

    if n < seq + 3:
        n = seq + 3
    d = tmp_path_factory.mktemp("tf")
    arr = np.random.default_rng(seed).integers(0, vocab, size=n,
                                               dtype=np.uint16)
    path = d / "c.bin"
    arr.tofile(path)
    cfg = load_config(None, [
        "model.name=gpt2", "model.vocab_size=65536",
        "data.kind=token_file", f"data.path={path}",
        f"data.batch_size={bsz}", f"data.seq_len={seq}",
        f"data.seed={seed}",
    ])
    data = TokenFileData(cfg, rank=0, device=torch.device("cpu"),
                         dtype=torch.float32)
    x, y = data.next_batch()
    assert x.shape == (bsz, seq) and y.shape == (bsz, seq)
    assert torch.equal(x[:, 1:], y[:, :-1])
    assert int(x.max()) < vocab and int(x.min()) >= 0


@settings(max_examples=100, deadline=None)
@given(
    lr=st.floats(1e-6, 1.0),
    warmup=st.integers(1, 50),
    seed=st.integers(0, 99),
)
def test_warmup_monotonic(lr, warmup, seed):
    cfg = load_config(None, [f"optim.lr={lr}",
                             f"optim.warmup_steps={warmup}",
                             "optim.schedule=cosine"])
    total = warmup + 50
    prev = 0.0
    for s_ in range(warmup):
        cur = lr_at(cfg, s_, total)
        assert cur >= prev - 1e-12
        prev = cur
    assert abs(lr_at(cfg, warmup - 1, total) - lr) < 1e-9
