"""ZeRO-1 optimizer-state sharding (dist.zero1): grads stay fully
all-reduced through the bucket engine, masters/moments are sharded,
owners broadcast updated params.  Must be TRAINING-EQUIVALENT to plain
DDP (fp32 CPU: bitwise), and checkpoint/resume/reshard-capable."""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from scaffold.config import load_config

WORLD = 2


def _cfg(zero1, ckpt_dir="", resume="", steps=5, dtype="fp32"):
    over = [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=2",
        "model.d_model=32", "model.vocab_size=64", "model.max_seq_len=16",
        "data.kind=synthetic_tokens", "data.batch_size=4", "data.seq_len=16",
        f"data.active_vocab=8", f"dtype={dtype}", "device=cpu",
        "log.interval=1000", "optim.lr=0.01", f"steps={steps}",
        f"dist.zero1={'true' if zero1 else 'false'}",
    ]
    if ckpt_dir:
        over += [f"ckpt.dir={ckpt_dir}", "ckpt.interval=100",
                 "ckpt.save_final=true"]
    if resume:
        over.append(f"ckpt.resume={resume}")
    return load_config(None, over)


def _worker(rank, world, store_path, zero1, ckpt_dir, resume, steps, q,
            dtype="fp32"):
    from scaffold.runtime import DistContext
    from scaffold.trainer import Trainer

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    ctx = DistContext(rank=rank, world_size=world, local_rank=rank,
                      backend="gloo", device=torch.device("cpu"),
                      initialized=True)
    t = Trainer(_cfg(zero1, ckpt_dir, resume, steps, dtype), ctx=ctx)
    t.fit()
    params = torch.cat([p.detach().flatten().float()
                        for p in t.raw_model.parameters()])
    own_master = sum(m.numel() for m in t.optimizer.master)
    total = sum(p.numel() for p in t.raw_model.parameters())
    if rank == 0:
        q.put((params.numpy(), own_master, total, t.step))
    dist.destroy_process_group()


def _run(zero1, tmp_path, tag, ckpt_dir="", resume="", steps=5,
         dtype="fp32"):
    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(
        target=_worker,
        args=(r, WORLD, str(tmp_path / f"store_{tag}"), zero1, ckpt_dir,
              resume, steps, q, dtype))
        for r in range(WORLD)]
    for p in procs:
        p.start()
    out = q.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    return out


@pytest.mark.timeout(300)
def test_zero1_matches_plain_ddp(tmp_path):
    p0, own0, total0, _ = _run(False, tmp_path, "ddp")
    p1, own1, total1, _ = _run(True, tmp_path, "z1")
    # optimizer really is sharded: rank 0 holds a strict subset
    assert own0 == total0
    assert 0 < own1 < total1
    # and training is exactly equivalent (fp32 CPU: bitwise)
    assert np.array_equal(p0, p1)


@pytest.mark.timeout(300)
def test_zero1_matches_plain_ddp_bf16(tmp_path):
    """bf16 working params (fp32 masters): still bitwise — owner-only
    arithmetic equals everyone's, broadcasts are lossless."""
    # bf16 eager CPU math is slow — 3 steps suffice for the claim
    p0, _, _, _ = _run(False, tmp_path, "ddpb", steps=3, dtype="bf16")
    p1, _, _, _ = _run(True, tmp_path, "z1b", steps=3, dtype="bf16")
    assert np.array_equal(p0, p1)


@pytest.mark.timeout(300)
def test_zero1_ckpt_resume_same_world(tmp_path):
    pa, _, _, _ = _run(True, tmp_path, "a", ckpt_dir=str(tmp_path / "ca"),
                       steps=6)
    # 4 steps, save, resume in a fresh world for 2 more
    _run(True, tmp_path, "b1", ckpt_dir=str(tmp_path / "cb"), steps=4)
    from scaffold.ckpt import layout
    mid = layout.latest_checkpoint(str(tmp_path / "cb"))
    pb, _, _, step = _run(True, tmp_path, "b2",
                          ckpt_dir=str(tmp_path / "cb2"), resume=mid,
                          steps=6)
    assert step == 6
    assert np.array_equal(pa, pb)


@pytest.mark.timeout(300)
def test_zero1_reshard_to_single_process(tmp_path):
    """zero1 ws=2 checkpoint -> ws=1 (full optimizer) load: the by-name
    path assembles the complete optimizer state from both shards."""
    from scaffold.trainer import Trainer

    _run(True, tmp_path, "r", ckpt_dir=str(tmp_path / "cr"), steps=4)
    from scaffold.ckpt import layout
    ck = layout.latest_checkpoint(str(tmp_path / "cr"))

    t = Trainer(_cfg(False, resume=ck, steps=4))
    assert t.step == 4
    # full optimizer state materialized (every moment nonzero-sized)
    assert sum(m.numel() for m in t.optimizer.master) == \
        sum(p.numel() for p in t.raw_model.parameters())
    assert t.optimizer.step_count == 4
    # moments carry real signal (loaded, not fresh zeros)
    assert any(m.abs().sum() > 0 for m in t.optimizer.exp_avg)


def _accum_worker(rank, world, store_path, zero1, accum, q):
    from scaffold.runtime import DistContext
    from scaffold.trainer import Trainer

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    ctx = DistContext(rank=rank, world_size=world, local_rank=rank,
                      backend="gloo", device=torch.device("cpu"),
                      initialized=True)
    cfg = _cfg(zero1, steps=4)
    cfg.grad_accum = accum
    t = Trainer(cfg, ctx=ctx)
    it = iter(t.data)
    for _ in range(4):
        t._train_step(it, accum)
        t.step += 1
    params = torch.cat([p.detach().flatten()
                        for p in t.raw_model.parameters()])
    if rank == 0:
        q.put(params.numpy())
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_zero1_with_grad_accumulation(tmp_path):
    """zero1 x grad_accum: broadcasts fire once per OPTIMIZER step, so
    the accumulated run must stay bitwise-equal to plain DDP's."""

    def run(zero1):
        ctxm = mp.get_context("spawn")
        q = ctxm.Queue()
        procs = [ctxm.Process(
            target=_accum_worker,
            args=(r, WORLD, str(tmp_path / f"s{zero1}"), zero1, 2, q))
            for r in range(WORLD)]
        for p in procs:
            p.start()
        out = q.get(timeout=200)
        for p in procs:
            p.join(timeout=60)
            assert p.exitcode == 0
        return out

    assert np.array_equal(run(False), run(True))


def _lora_worker(rank, world, store_path, zero1, q):
    from scaffold.runtime import DistContext
    from scaffold.trainer import Trainer

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    ctx = DistContext(rank=rank, world_size=world, local_rank=rank,
                      backend="gloo", device=torch.device("cpu"),
                      initialized=True)
    cfg = _cfg(zero1, steps=4)
    cfg.model.lora_r = 4
    t = Trainer(cfg, ctx=ctx)
    t.fit()
    params = torch.cat([p.detach().flatten().float()
                        for p in t.raw_model.parameters()])
    n_opt = len(t.optimizer.params)
    n_trainable = sum(1 for p in t.raw_model.parameters() if p.requires_grad)
    if rank == 0:
        q.put((params.numpy(), n_opt, n_trainable))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_zero1_composes_with_lora(tmp_path):
    """ZeRO-1 over a LoRA model shards only the ADAPTER params and
    stays bitwise-equivalent to plain DDP LoRA."""

    def run(zero1, tag):
        ctxm = mp.get_context("spawn")
        q = ctxm.Queue()
        procs = [ctxm.Process(target=_lora_worker,
                              args=(r, WORLD, str(tmp_path / tag), zero1, q))
                 for r in range(WORLD)]
        for p in procs:
            p.start()
        out = q.get(timeout=200)
        for p in procs:
            p.join(timeout=60)
            assert p.exitcode == 0
        return out

    p0, n0, tr0 = run(False, "l_ddp")
    p1, n1, tr1 = run(True, "l_z1")
    assert n0 == tr0                    # plain: optimizer covers all adapters
    assert 0 < n1 < tr1                 # zero1: a strict shard of them
    assert np.array_equal(p0, p1)
