"""Multi-process DDP equivalence on CPU/gloo: N ranks with
BucketedDDP produce grads identical to a single-process large-batch run."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from scaffold.config import load_config
from scaffold.runtime import DistContext

WORLD = 2


def _cfg(batch_size):
    return load_config(None, [
        "model.name=mlp", "model.in_dim=64", "model.hidden_dim=128",
        "model.out_dim=8", "data.kind=synthetic_vector",
        f"data.batch_size={batch_size}", "data.vector_dim=64",
        "dtype=fp32", "device=cpu", "log.interval=1000",
        "dist.bucket_cap_mb=0.125", "dist.first_bucket_mb=0.03",
    ])


def _worker(rank, world, file_store_path, result_q):
    from scaffold.trainer import Trainer

    store = dist.FileStore(file_store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    cfg = _cfg(batch_size=8)
    ctx = DistContext(rank=rank, world_size=world, local_rank=rank,
                      backend="gloo", device=torch.device("cpu"), initialized=True)
    t = Trainer(cfg, ctx=ctx)

    # fixed per-rank batch: deterministic slices of one global batch
    g = torch.Generator().manual_seed(42)
    X = torch.randn(world * 8, 64, generator=g)
    Y = torch.randn(world * 8, 8, generator=g)
    x = X[rank * 8:(rank + 1) * 8]
    y = Y[rank * 8:(rank + 1) * 8]

    t.ddp.zero_grad()
    loss = t.model(x, y)
    loss.backward()
    t.ddp.finish()
    scale = t.ddp.grad_scale
    grads = torch.cat([p.grad.flatten() * scale for p in t.raw_model.parameters()])
    params = torch.cat([p.detach().flatten() for p in t.raw_model.parameters()])
    if rank == 0:
        # numpy arrays pickle by value; torch tensors over mp.Queue use fd
        # passing and break if the worker exits before the parent reads
        result_q.put(("dist", grads.numpy(), params.numpy()))
    dist.destroy_process_group()


@pytest.mark.timeout(240)
@pytest.mark.parametrize("world", [2, 4])
def test_ddp_matches_large_batch(tmp_path, world):
    from scaffold.trainer import Trainer

    ctxmp = mp.get_context("spawn")
    q = ctxmp.Queue()
    store_path = str(tmp_path / "filestore")
    procs = [
        ctxmp.Process(target=_worker, args=(r, world, store_path, q))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    kind, dist_grads, dist_params = q.get(timeout=100)
    dist_grads = torch.from_numpy(dist_grads)
    dist_params = torch.from_numpy(dist_params)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # single-process reference: same model seed, the whole global batch
    cfg = _cfg(batch_size=world * 8)
    t = Trainer(cfg)
    ref_params = torch.cat([p.detach().flatten() for p in t.raw_model.parameters()])
    assert torch.allclose(ref_params, dist_params), "init params differ (broadcast/seed)"

    g = torch.Generator().manual_seed(42)
    X = torch.randn(world * 8, 64, generator=g)
    Y = torch.randn(world * 8, 8, generator=g)
    t.ddp.zero_grad()
    t.model(X, Y).backward()
    t.ddp.finish()
    ref_grads = torch.cat([p.grad.flatten() for p in t.raw_model.parameters()])

    # mean-loss over the global batch == average of per-rank mean losses
    assert torch.allclose(ref_grads, dist_grads, rtol=1e-5, atol=1e-7), (
        (ref_grads - dist_grads).abs().max().item()
    )


def _worker_full_fit(rank, world, file_store_path, ckpt_dir, result_q):
    """The FULL fit loop at ws=2: eval cadence + checkpoint cadence +
    metric logging — the paths that only fire on real multi-rank runs
    (evaluate()'s all_reduce had never executed at ws>1)."""
    from scaffold.trainer import Trainer

    store = dist.FileStore(file_store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    cfg = _cfg(batch_size=8)
    cfg.steps = 6
    cfg.log.interval = 2
    cfg.log.eval_interval = 3
    cfg.log.eval_steps = 2
    cfg.ckpt.interval = 3
    cfg.ckpt.dir = ckpt_dir
    cfg.ckpt.save_final = True
    ctx = DistContext(rank=rank, world_size=world, local_rank=rank,
                      backend="gloo", device=torch.device("cpu"), initialized=True)
    t = Trainer(cfg, ctx=ctx)
    t.fit()
    ev = t.evaluate(eval_steps=2)
    if rank == 0:
        result_q.put(("ok", float(ev), t.step))
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_full_fit_eval_ckpt_ws2(tmp_path):
    ctxmp = mp.get_context("spawn")
    q = ctxmp.Queue()
    ckpt_dir = str(tmp_path / "ckpt")
    procs = [
        ctxmp.Process(target=_worker_full_fit,
                      args=(r, WORLD, str(tmp_path / "fs_full"), ckpt_dir, q))
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    kind, ev, step = q.get(timeout=150)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert kind == "ok"
    assert torch.isfinite(torch.tensor(ev))
    assert step == 6
    # both cadence checkpoint and final checkpoint committed
    assert os.path.isdir(os.path.join(ckpt_dir, "step_00000003"))
    assert os.path.isdir(os.path.join(ckpt_dir, "step_00000006"))


def _worker_accum(rank, world, file_store_path, result_q):
    from scaffold.trainer import Trainer

    store = dist.FileStore(file_store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    cfg = _cfg(batch_size=4)
    cfg.grad_accum = 2
    ctx = DistContext(rank=rank, world_size=world, local_rank=rank,
                      backend="gloo", device=torch.device("cpu"), initialized=True)
    t = Trainer(cfg, ctx=ctx)

    # 4 deterministic micro-batches = one global batch of 16
    g = torch.Generator().manual_seed(77)
    X = torch.randn(world * 2 * 4, 64, generator=g)
    Y = torch.randn(world * 2 * 4, 8, generator=g)
    micros = [(X[i * 4:(i + 1) * 4], Y[i * 4:(i + 1) * 4])
              for i in range(world * 2)]
    mine = iter([micros[rank], micros[world + rank]])

    t._train_step(mine, accum=2)
    params = torch.cat([p.detach().flatten() for p in t.raw_model.parameters()])
    if rank == 0:
        result_q.put(params.numpy())
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_grad_accum_matches_large_batch(tmp_path):
    """no_sync micro-steps + one reduced step == single-process batch 16."""
    from scaffold.trainer import Trainer

    ctxmp = mp.get_context("spawn")
    q = ctxmp.Queue()
    procs = [
        ctxmp.Process(target=_worker_accum, args=(r, WORLD, str(tmp_path / "fs"), q))
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    dist_params = torch.from_numpy(q.get(timeout=100))
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    cfg = _cfg(batch_size=WORLD * 2 * 4)
    t = Trainer(cfg)
    g = torch.Generator().manual_seed(77)
    X = torch.randn(WORLD * 2 * 4, 64, generator=g)
    Y = torch.randn(WORLD * 2 * 4, 8, generator=g)
    t._train_step(iter([(X, Y)]), accum=1)
    ref_params = torch.cat([p.detach().flatten() for p in t.raw_model.parameters()])
    assert torch.allclose(ref_params, dist_params, rtol=1e-5, atol=1e-6), (
        (ref_params - dist_params).abs().max().item()
    )
