"""Context-parallel attention equivalence (SURVEY §5.7 seam): ws=2
sequence-sharded cp_attention == single-process full attention, for the
output AND all input gradients (fp32, CPU/gloo)."""

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from scaffold import ops

WORLD = 2
B, H, HKV, S, D = 2, 4, 2, 64, 32  # S_total = 64 -> 32 per rank


def _make_inputs():
    g = torch.Generator().manual_seed(123)
    q = torch.randn(B, H, S, D, generator=g)
    k = torch.randn(B, HKV, S, D, generator=g)
    v = torch.randn(B, HKV, S, D, generator=g)
    do = torch.randn(B, H, S, D, generator=g)
    return q, k, v, do


def _worker(rank, world, store_path, causal, q_ret, variant="allgather"):
    from scaffold.dist.cp import cp_attention, cp_attention_ring

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    q, k, v, do = _make_inputs()
    sl = S // world
    qs = q[:, :, rank * sl:(rank + 1) * sl].clone().requires_grad_(True)
    ks = k[:, :, rank * sl:(rank + 1) * sl].clone().requires_grad_(True)
    vs = v[:, :, rank * sl:(rank + 1) * sl].clone().requires_grad_(True)
    fn = cp_attention_ring if variant == "ring" else cp_attention
    o = fn(qs, ks, vs, rank, world, causal=causal)
    o.backward(do[:, :, rank * sl:(rank + 1) * sl])
    q_ret.put((rank, o.detach().numpy(), qs.grad.numpy(),
               ks.grad.numpy(), vs.grad.numpy()))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
@pytest.mark.parametrize("variant", ["allgather", "ring"])
@pytest.mark.parametrize("causal", [True, False])
def test_cp_matches_full_attention(tmp_path, causal, variant):
    ctxmp = mp.get_context("spawn")
    ret = ctxmp.Queue()
    procs = [
        ctxmp.Process(target=_worker,
                      args=(r, WORLD, str(tmp_path / f"fs{causal}{variant}"),
                            causal, ret, variant))
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    shards = {}
    for _ in range(WORLD):
        r, o, dq, dk, dv = ret.get(timeout=120)
        shards[r] = (torch.from_numpy(o), torch.from_numpy(dq),
                     torch.from_numpy(dk), torch.from_numpy(dv))
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # single-process reference on the FULL sequence
    q, k, v, do = _make_inputs()
    q = q.requires_grad_(True)
    k = k.requires_grad_(True)
    v = v.requires_grad_(True)
    o_ref, _ = ops._eager_attn_lse(q, k, v, causal)
    o_ref.backward(do)

    sl = S // WORLD
    o_cat = torch.cat([shards[r][0] for r in range(WORLD)], dim=2)
    dq_cat = torch.cat([shards[r][1] for r in range(WORLD)], dim=2)
    dk_cat = torch.cat([shards[r][2] for r in range(WORLD)], dim=2)
    dv_cat = torch.cat([shards[r][3] for r in range(WORLD)], dim=2)
    for name, got, want in [
        ("o", o_cat, o_ref.detach()),
        ("dq", dq_cat, q.grad),
        ("dk", dk_cat, k.grad),
        ("dv", dv_cat, v.grad),
    ]:
        err = (got - want).abs().max().item()
        assert err < 2e-5 * max(1.0, want.abs().max().item()), f"{name}: {err}"


def _worker_ulysses(rank, world, store_path, causal, q_ret):
    from scaffold.dist.cp import ulysses_attention

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    q, k, v, do = _make_inputs()
    sl = S // world
    qs = q[:, :, rank * sl:(rank + 1) * sl].clone().requires_grad_(True)
    ks = k[:, :, rank * sl:(rank + 1) * sl].clone().requires_grad_(True)
    vs = v[:, :, rank * sl:(rank + 1) * sl].clone().requires_grad_(True)
    o = ulysses_attention(qs, ks, vs, rank, world, causal=causal)
    o.backward(do[:, :, rank * sl:(rank + 1) * sl])
    q_ret.put((rank, o.detach().numpy(), qs.grad.numpy(),
               ks.grad.numpy(), vs.grad.numpy()))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
@pytest.mark.parametrize("causal", [True, False])
def test_ulysses_matches_full_attention(tmp_path, causal):
    ctxmp = mp.get_context("spawn")
    ret = ctxmp.Queue()
    procs = [
        ctxmp.Process(target=_worker_ulysses,
                      args=(r, WORLD, str(tmp_path / f"ufs{causal}"), causal, ret))
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    shards = {}
    for _ in range(WORLD):
        r, o, dq, dk, dv = ret.get(timeout=120)
        shards[r] = (torch.from_numpy(o), torch.from_numpy(dq),
                     torch.from_numpy(dk), torch.from_numpy(dv))
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    q, k, v, do = _make_inputs()
    q = q.requires_grad_(True)
    k = k.requires_grad_(True)
    v = v.requires_grad_(True)
    o_ref, _ = ops._eager_attn_lse(q, k, v, causal)
    o_ref.backward(do)

    o_cat = torch.cat([shards[r][0] for r in range(WORLD)], dim=2)
    dq_cat = torch.cat([shards[r][1] for r in range(WORLD)], dim=2)
    dk_cat = torch.cat([shards[r][2] for r in range(WORLD)], dim=2)
    dv_cat = torch.cat([shards[r][3] for r in range(WORLD)], dim=2)
    for name, got, want in [
        ("o", o_cat, o_ref.detach()),
        ("dq", dq_cat, q.grad),
        ("dk", dk_cat, k.grad),
        ("dv", dv_cat, v.grad),
    ]:
        err = (got - want).abs().max().item()
        assert err < 2e-5 * max(1.0, want.abs().max().item()), f"{name}: {err}"
