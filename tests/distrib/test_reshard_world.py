"""Cross-world-size checkpoint reshard: shards written by
a 2-rank world load by-name into a single-process trainer, bit-exact."""

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

import pytest

from scaffold.config import load_config
from scaffold.runtime import DistContext

WORLD = 2


def _cfg(ckpt_dir, resume=""):
    over = [
        "model.name=mlp", "model.in_dim=32", "model.hidden_dim=64",
        "model.out_dim=4", "data.kind=synthetic_vector",
        "data.batch_size=8", "data.vector_dim=32",
        "dtype=fp32", "device=cpu", "log.interval=1000",
        f"ckpt.dir={ckpt_dir}",
    ]
    if resume:
        over.append(f"ckpt.resume={resume}")
    return load_config(None, over)


def _worker(rank, world, store_path, ckpt_dir, q):
    from scaffold.trainer import Trainer

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    ctx = DistContext(rank=rank, world_size=world, local_rank=rank,
                      backend="gloo", device=torch.device("cpu"), initialized=True)
    t = Trainer(_cfg(ckpt_dir), ctx=ctx)
    it = iter(t.data)
    for _ in range(3):
        t._train_step(it, 1)
    t.step = 3
    path = t.save_checkpoint()
    if rank == 0:
        params = torch.cat([p.detach().flatten() for p in t.raw_model.parameters()])
        moms = torch.cat([m.flatten() for m in t.optimizer.exp_avg])
        q.put((path, params.numpy(), moms.numpy()))
    dist.destroy_process_group()


def _worker_stream(rank, world, store_path, ckpt_dir, q):
    """Phase 2: a FRESH ws=2 world resumes the ws=2 checkpoint — the
    same-world STREAMING load path (own-shard read + owner broadcasts)."""
    from scaffold.trainer import Trainer
    from scaffold.ckpt import layout

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    ctx = DistContext(rank=rank, world_size=world, local_rank=rank,
                      backend="gloo", device=torch.device("cpu"), initialized=True)
    path = layout.latest_checkpoint(ckpt_dir)
    cfg = _cfg(ckpt_dir, resume=path)
    cfg.seed = 777  # different init: the load must overwrite everything
    t = Trainer(cfg, ctx=ctx)
    params = torch.cat([p.detach().flatten() for p in t.raw_model.parameters()])
    moms = torch.cat([m.flatten() for m in t.optimizer.exp_avg])
    q.put((rank, t.step, params.numpy(), moms.numpy()))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(240)
def test_ws2_streaming_resume_matches(tmp_path):
    """ws=2 save -> ws=2 resume through the streaming path: every rank's
    full state bit-matches the saved rank-0 state."""
    ctxmp = mp.get_context("spawn")
    q = ctxmp.Queue()
    procs = [
        ctxmp.Process(target=_worker,
                      args=(r, WORLD, str(tmp_path / "fs1"), str(tmp_path / "ck"), q))
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    _path, params_saved, moms_saved = q.get(timeout=120)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    q2 = ctxmp.Queue()
    procs = [
        ctxmp.Process(target=_worker_stream,
                      args=(r, WORLD, str(tmp_path / "fs2"), str(tmp_path / "ck"), q2))
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    results = [q2.get(timeout=120) for _ in range(WORLD)]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    for rank, step, params, moms in results:
        assert step == 3
        assert torch.equal(torch.from_numpy(params), torch.from_numpy(params_saved)), rank
        assert torch.equal(torch.from_numpy(moms), torch.from_numpy(moms_saved)), rank


@pytest.mark.timeout(180)
def test_ws2_checkpoint_loads_into_ws1(tmp_path):
    from scaffold.trainer import Trainer

    ctxmp = mp.get_context("spawn")
    q = ctxmp.Queue()
    procs = [
        ctxmp.Process(target=_worker,
                      args=(r, WORLD, str(tmp_path / "fs"), str(tmp_path / "ck"), q))
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    path, params2, moms2 = q.get(timeout=120)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    cfg = _cfg(str(tmp_path / "ck"), resume=path)
    cfg.seed = 4242  # different init — resume must overwrite everything
    t1 = Trainer(cfg)
    assert t1.step == 3
    params1 = torch.cat([p.detach().flatten() for p in t1.raw_model.parameters()])
    moms1 = torch.cat([m.flatten() for m in t1.optimizer.exp_avg])
    assert torch.equal(params1, torch.from_numpy(params2))
    assert torch.equal(moms1, torch.from_numpy(moms2))
