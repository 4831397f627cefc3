"""Expert-parallel MoE equivalence: ws=2 expert-sharded
top-1 MoE == single-process all-experts MoE for output, dx, expert
weight grads, and (rank-summed, as DDP would) gate grads."""

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2
T, DM, DF, E = 8, 16, 32, 4  # T tokens per rank


def _seed_weights(moe, base):
    g = torch.Generator().manual_seed(777)
    up_rows = moe.w_up[0].shape[0]        # DF (gelu) or 2*DF (swiglu)
    gate = torch.randn(E, DM, generator=g) * 0.3
    ups = [torch.randn(up_rows, DM, generator=g) * 0.1 for _ in range(E)]
    downs = [torch.randn(DM, DF, generator=g) * 0.1 for _ in range(E)]
    X = torch.randn(WORLD * T, DM, generator=g)
    dy = torch.randn(WORLD * T, DM, generator=g)
    with torch.no_grad():
        moe.gate.weight.copy_(gate)
        for i in range(len(moe.w_up)):
            moe.w_up[i].copy_(ups[base + i])
            moe.w_down[i].copy_(downs[base + i])
    return X, dy


def _worker(rank, world, store_path, q_ret, dispatch="dense", top_k=1,
            act="gelu"):
    from scaffold.dist.ep import ExpertParallelMoE

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    moe = ExpertParallelMoE(DM, DF, E, world, rank, dispatch=dispatch,
                            top_k=top_k, act=act)
    X, dy = _seed_weights(moe, rank * (E // world))
    xl = X[rank * T:(rank + 1) * T].clone().requires_grad_(True)
    y = moe(xl)
    y.backward(dy[rank * T:(rank + 1) * T])
    q_ret.put((rank, y.detach().numpy(), xl.grad.numpy(),
               moe.gate.weight.grad.numpy(),
               [w.grad.numpy() for w in moe.w_up],
               [w.grad.numpy() for w in moe.w_down]))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
@pytest.mark.parametrize("dispatch", ["dense", "alltoall"])
@pytest.mark.parametrize("top_k,act", [(1, "gelu"), (2, "gelu"),
                                       (2, "swiglu")])
def test_ep_moe_matches_full(tmp_path, dispatch, top_k, act):
    from scaffold.dist.ep import ExpertParallelMoE

    ctxmp = mp.get_context("spawn")
    ret = ctxmp.Queue()
    procs = [ctxmp.Process(
        target=_worker,
        args=(r, WORLD, str(tmp_path / f"fs{dispatch}{top_k}{act}"), ret,
              dispatch, top_k, act))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    shards = {}
    for _ in range(WORLD):
        r, *rest = ret.get(timeout=120)
        shards[r] = rest
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # single-process reference: world=1 holds ALL experts
    ref = ExpertParallelMoE(DM, DF, E, 1, 0, top_k=top_k, act=act)
    X, dy = _seed_weights(ref, 0)
    x = X.clone().requires_grad_(True)
    y_ref = ref(x)
    y_ref.backward(dy)

    y_cat = torch.cat([torch.from_numpy(shards[r][0]) for r in range(WORLD)])
    dx_cat = torch.cat([torch.from_numpy(shards[r][1]) for r in range(WORLD)])
    assert torch.allclose(y_cat, y_ref.detach(), atol=2e-5)
    assert torch.allclose(dx_cat, x.grad, atol=2e-5)

    # gate is replicated: DDP would SUM(avg) grads — compare the sum
    gate_sum = sum(torch.from_numpy(shards[r][2]) for r in range(WORLD))
    assert torch.allclose(gate_sum, ref.gate.weight.grad, atol=2e-5)

    # expert weights are rank-local shards of the full expert set
    el = E // WORLD
    for r in range(WORLD):
        for i in range(el):
            assert torch.allclose(torch.from_numpy(shards[r][3][i]),
                                  ref.w_up[r * el + i].grad, atol=2e-5)
            assert torch.allclose(torch.from_numpy(shards[r][4][i]),
                                  ref.w_down[r * el + i].grad, atol=2e-5)
