"""Fake-backend tier: verify bucket schedule, overlap
ordering and message sizes with zero GPUs and zero extra processes."""

import torch

from scaffold.config import load_config
from scaffold.dist.fake_pg import RecordingComm
from scaffold.trainer import Trainer


def _cfg():
    return load_config(None, [
        "model.name=mlp", "model.in_dim=256", "model.hidden_dim=512",
        "data.kind=synthetic_vector", "data.batch_size=8",
        "dtype=fp32", "device=cpu", "steps=2", "log.interval=1000",
        "dist.bucket_cap_mb=1", "dist.first_bucket_mb=0.25",
    ])


def test_bucket_allreduce_schedule():
    comm = RecordingComm(world_size=8)
    t = Trainer(_cfg(), comm=comm)
    n_buckets = len(t.ddp.buckets)
    assert n_buckets >= 2  # MLP ~ 270k params fp32 > 1 MiB

    bcasts = comm.ops("broadcast")
    assert len(bcasts) == len(list(t.raw_model.parameters()))

    x, y = t.data.next_batch()
    loss = t.model(x, y)
    loss.backward()
    t.ddp.finish()

    ars = comm.ops("all_reduce")
    assert len(ars) == n_buckets
    # every bucket all-reduce is async (overlap with backward)
    assert all(r.async_op for r in ars)
    # total reduced bytes == total grad bytes
    total = sum(p.numel() * 4 for p in t.raw_model.parameters())
    assert sum(r.nbytes for r in ars) == total
    # overlap ordering: all-reduces launch in increasing order during bwd
    orders = [r.order for r in ars]
    assert orders == sorted(orders)


def test_second_step_relaunches():
    comm = RecordingComm(world_size=2)
    t = Trainer(_cfg(), comm=comm)
    for _ in range(2):
        t.ddp.zero_grad()
        x, y = t.data.next_batch()
        t.model(x, y).backward()
        t.ddp.finish()
    assert len(comm.ops("all_reduce")) == 2 * len(t.ddp.buckets)


def test_no_sync_skips_reduction():
    comm = RecordingComm(world_size=2)
    t = Trainer(_cfg(), comm=comm)
    x, y = t.data.next_batch()
    with t.ddp.no_sync():
        t.model(x, y).backward()
        t.ddp.finish()
    assert len(comm.ops("all_reduce")) == 0
    t.model(x, y).backward()
    t.ddp.finish()
    assert len(comm.ops("all_reduce")) == len(t.ddp.buckets)


def test_grad_views_share_storage():
    comm = RecordingComm(world_size=2)
    t = Trainer(_cfg(), comm=comm)
    for b in t.ddp.buckets:
        for p, v in zip(b.params, b.views):
            assert p.grad is not None
            assert p.grad.data_ptr() >= b.flat.data_ptr()
            assert p.grad.untyped_storage().data_ptr() == b.flat.untyped_storage().data_ptr()
