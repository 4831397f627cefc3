"""Tensor-parallel primitives equivalence (SURVEY §2.4 additive layer):
a ws=2 column->row parallel MLP matches the single-process full MLP for
output AND gradients (activation grads bitwise-mergeable; weight grads
are the reference's shards)."""

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2
B, DM, DF = 4, 32, 64


def _full_weights():
    g = torch.Generator().manual_seed(321)
    w1 = torch.randn(DF, DM, generator=g) * 0.1
    b1 = torch.randn(DF, generator=g) * 0.1
    w2 = torch.randn(DM, DF, generator=g) * 0.1
    b2 = torch.randn(DM, generator=g) * 0.1
    x = torch.randn(B, DM, generator=g)
    dy = torch.randn(B, DM, generator=g)
    return w1, b1, w2, b2, x, dy


def _worker(rank, world, store_path, q_ret):
    from scaffold.dist.tp import TpMLP

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    w1, b1, w2, b2, x, dy = _full_weights()
    m = TpMLP(DM, DF, world, rank)
    m.fc.load_full(w1, b1)
    m.proj.load_full(w2, b2)
    xl = x.clone().requires_grad_(True)
    y = m(xl)
    y.backward(dy)
    q_ret.put((rank, y.detach().numpy(), xl.grad.numpy(),
               m.fc.weight.grad.numpy(), m.proj.weight.grad.numpy(),
               m.fc.bias.grad.numpy()))
    dist.barrier()
    dist.destroy_process_group()


def _worker_hybrid(rank, world, store_path, q_ret):
    """TP degree 1 inside a ws=2 world (pure-DP corner of hybrid TP x DP):
    each rank runs an unsharded TpMLP over its own single-rank TP group.
    dx must equal the single-process dx — the r1 bug (group=None in
    _CopyToTp) all-reduced dx over WORLD and doubled it."""
    from scaffold.dist.tp import TpMLP

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    # every rank must create every group (new_group is collective)
    groups = [dist.new_group([r]) for r in range(world)]
    w1, b1, w2, b2, x, dy = _full_weights()
    m = TpMLP(DM, DF, 1, 0, group=groups[rank])
    m.fc.load_full(w1, b1)
    m.proj.load_full(w2, b2)
    xl = x.clone().requires_grad_(True)
    y = m(xl)
    y.backward(dy)
    q_ret.put((rank, xl.grad.numpy(), m.fc.weight.grad.numpy()))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_tp_subgroup_hybrid_dp(tmp_path):
    ctxmp = mp.get_context("spawn")
    ret = ctxmp.Queue()
    procs = [ctxmp.Process(target=_worker_hybrid,
                           args=(r, WORLD, str(tmp_path / "fs2"), ret))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    shards = {}
    for _ in range(WORLD):
        r, dx, dw1 = ret.get(timeout=120)
        shards[r] = (torch.from_numpy(dx), torch.from_numpy(dw1))
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    w1, b1, w2, b2, x, dy = _full_weights()
    w1 = w1.requires_grad_(True)
    x = x.requires_grad_(True)
    y_ref = torch.nn.functional.gelu(
        torch.nn.functional.linear(x, w1, b1), approximate="tanh") @ w2.t() + b2
    y_ref.backward(dy)
    for r in range(WORLD):
        dx, dw1 = shards[r]
        assert torch.allclose(dx, x.grad, atol=2e-5), (
            "dx wrong under TP subgroup (was it reduced over WORLD?)")
        assert torch.allclose(dw1, w1.grad, atol=2e-5)


@pytest.mark.timeout(180)
def test_tp_mlp_matches_full(tmp_path):
    ctxmp = mp.get_context("spawn")
    ret = ctxmp.Queue()
    procs = [ctxmp.Process(target=_worker, args=(r, WORLD, str(tmp_path / "fs"), ret))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    shards = {}
    for _ in range(WORLD):
        r, y, dx, dw1, dw2, db1 = ret.get(timeout=120)
        shards[r] = tuple(torch.from_numpy(a) for a in (y, dx, dw1, dw2, db1))
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # single-process reference
    w1, b1, w2, b2, x, dy = _full_weights()
    w1 = w1.requires_grad_(True)
    b1 = b1.requires_grad_(True)
    w2 = w2.requires_grad_(True)
    x = x.requires_grad_(True)
    y_ref = torch.nn.functional.gelu(
        torch.nn.functional.linear(x, w1, b1), approximate="tanh") @ w2.t() + b2
    y_ref.backward(dy)

    half = DF // WORLD
    for r in range(WORLD):
        y, dx, dw1, dw2, db1 = shards[r]
        assert torch.allclose(y, y_ref.detach(), atol=2e-5), r
        assert torch.allclose(dx, x.grad, atol=2e-5), r
        assert torch.allclose(dw1, w1.grad[r * half:(r + 1) * half], atol=2e-5)
        assert torch.allclose(db1, b1.grad[r * half:(r + 1) * half], atol=2e-5)
        assert torch.allclose(dw2, w2.grad[:, r * half:(r + 1) * half], atol=2e-5)


def _tp_decode_worker(rank, world, store_path, q):
    import torch.distributed as dist

    from scaffold.config import load_config
    from scaffold.dist.tp import shard_llama_tp
    from scaffold.generate import Generator
    from scaffold.models import build_model

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
        "model.vocab_size=128", "model.max_seq_len=64",
    ])
    torch.manual_seed(91)
    m = build_model(cfg).eval()
    shard_llama_tp(m, rank, world)
    # per-rank weight footprint of the sharded blocks really halves
    blk_numel = sum(p.numel() for b in m.blocks for p in b.parameters())
    prompt = torch.arange(1, 7).view(1, 6)
    out = Generator(m).generate(prompt, 16)
    # continuous-batching decode path (varpos) through the same shards
    from scaffold.serve import ContinuousBatcher
    cb = ContinuousBatcher(m, max_slots=2)
    try:
        r = cb.submit(prompt[0].tolist(), 16)
        assert r.event.wait(120) and r.error is None
        cb_toks = r.result
    finally:
        cb.close()
    if rank == 0:
        q.put((out.numpy(), cb_toks, blk_numel))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_tp_sharded_decode_matches_full(tmp_path):
    """TP=2 inference: sharded heads + gate|up with generator-side
    all_reduces reproduce the unsharded rollout token-for-token, on
    both the prefill/decode path and the continuous-batching varpos
    path; per-rank block weights actually halve."""
    import numpy as np

    from scaffold.config import load_config
    from scaffold.generate import Generator
    from scaffold.models import build_model

    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_tp_decode_worker,
                          args=(r, 2, str(tmp_path / "tpd"), q))
             for r in range(2)]
    for p in procs:
        p.start()
    out_tp, cb_toks, blk_numel = q.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
        "model.vocab_size=128", "model.max_seq_len=64",
    ])
    torch.manual_seed(91)
    full = build_model(cfg).eval()
    prompt = torch.arange(1, 7).view(1, 6)
    ref = Generator(full).generate(prompt, 16)
    assert np.array_equal(out_tp, ref.numpy())
    assert cb_toks == ref[0, 6:].tolist()
    full_blk = sum(p.numel() for b in full.blocks for p in b.parameters())
    norms = sum(p.numel() for b in full.blocks
                for mod in (b.attn_norm, b.mlp_norm)
                for p in mod.parameters())
    # sharded = replicated norms + exactly half of the projections
    assert blk_numel == norms + (full_blk - norms) // 2


def _tp_gpt2_worker(rank, world, store_path, q):
    import torch.distributed as dist

    from scaffold.config import load_config
    from scaffold.dist.tp import shard_llama_tp
    from scaffold.generate import Generator
    from scaffold.models import build_model

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=64", "model.vocab_size=128", "model.max_seq_len=64",
    ])
    torch.manual_seed(93)
    m = build_model(cfg).eval()
    shard_llama_tp(m, rank, world)   # dispatches to the gpt2 variant
    out = Generator(m).generate(torch.arange(1, 6).view(1, 5), 12)
    if rank == 0:
        q.put(out.numpy())
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_tp_gpt2_decode_matches_full(tmp_path):
    """GPT-2 TP=2: packed-qkv bias shards column-parallel, row-parallel
    biases add AFTER the reduce; sharded rollout == unsharded."""
    import numpy as np

    from scaffold.config import load_config
    from scaffold.generate import Generator
    from scaffold.models import build_model

    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_tp_gpt2_worker,
                          args=(r, 2, str(tmp_path / "tpg"), q))
             for r in range(2)]
    for p in procs:
        p.start()
    out_tp = q.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=64", "model.vocab_size=128", "model.max_seq_len=64",
    ])
    torch.manual_seed(93)
    full = build_model(cfg).eval()
    ref = Generator(full).generate(torch.arange(1, 6).view(1, 5), 12)
    assert np.array_equal(out_tp, ref.numpy())


def test_graphed_paths_guard_tp(tmp_path):
    """A TP-sharded model must FAIL LOUDLY on generate_graphed (the
    static step has no reductions) instead of emitting wrong tokens."""
    import torch.distributed as dist

    from scaffold.config import load_config
    from scaffold.dist.tp import shard_llama_tp
    from scaffold.generate import Generator
    from scaffold.models import build_model

    store = dist.FileStore(str(tmp_path / "fs"), 1)
    dist.init_process_group("gloo", store=store, rank=0, world_size=1)
    try:
        cfg = load_config(None, [
            "model.name=llama", "model.n_layers=1", "model.n_heads=4",
            "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
            "model.vocab_size=64", "model.max_seq_len=32",
        ])
        m = build_model(cfg).eval()
        shard_llama_tp(m, 0, 2)       # markers set (world=2 semantics)
        with pytest.raises(NotImplementedError, match="TP"):
            Generator(m).generate_graphed(torch.tensor([[1, 2]]), 4)
    finally:
        dist.destroy_process_group()
