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
