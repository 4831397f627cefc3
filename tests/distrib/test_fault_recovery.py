"""Failure-detection / elastic-recovery tier: a rank
is killed mid-run; the relaunched world resumes from the last COMPLETE
checkpoint manifest and continues bit-exact with an uninterrupted run.

The recovery model (deliberately minimal, §5.3): manifests are written
atomically (shards first, fsync, rank-0 manifest last), so the last
step with a manifest on disk is always consistent; the launcher's job
is only to restart the world with ckpt.resume pointing there.
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from scaffold.ckpt import layout
from scaffold.config import load_config
from scaffold.runtime import DistContext

WORLD = 2


def _cfg(ckpt_dir, resume=""):
    over = [
        "model.name=mlp", "model.in_dim=32", "model.hidden_dim=64",
        "model.out_dim=4", "data.kind=synthetic_vector",
        "data.batch_size=8", "data.vector_dim=32",
        "dtype=fp32", "device=cpu", "log.interval=1000",
        f"ckpt.dir={ckpt_dir}", "ckpt.interval=2", "ckpt.save_final=false",
        "steps=6",
    ]
    if resume:
        over.append(f"ckpt.resume={resume}")
    return load_config(None, over)


def _worker(rank, world, store_path, ckpt_dir, die_at_step, resume, q):
    from scaffold.trainer import Trainer

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    ctx = DistContext(rank=rank, world_size=world, local_rank=rank,
                      backend="gloo", device=torch.device("cpu"), initialized=True)
    t = Trainer(_cfg(ckpt_dir, resume), ctx=ctx)
    it = iter(t.data)
    while t.step < t.cfg.steps:
        if die_at_step is not None and t.step == die_at_step and rank == 1:
            os._exit(17)  # fault injection: rank dies without cleanup
        t._train_step(it, 1)
        t.step += 1
        if t.step % t.cfg.ckpt.interval == 0:
            t.save_checkpoint()
    if rank == 0:
        params = torch.cat([p.detach().flatten() for p in t.raw_model.parameters()])
        q.put((t.step, params.numpy()))
    dist.destroy_process_group()


def _launch(store_path, ckpt_dir, die_at_step, resume, q):
    ctxmp = mp.get_context("spawn")
    procs = [
        ctxmp.Process(target=_worker,
                      args=(r, WORLD, store_path, ckpt_dir, die_at_step, resume, q))
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    return procs


@pytest.mark.timeout(300)
def test_rank_failure_resume_matches_uninterrupted(tmp_path):
    ctxmp = mp.get_context("spawn")

    # ---- run A: uninterrupted 6 steps (the reference)
    qa = ctxmp.Queue()
    ckpt_a = str(tmp_path / "a")
    procs = _launch(str(tmp_path / "store_a"), ckpt_a, None, "", qa)
    step_a, params_a = qa.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # ---- run B: rank 1 dies at step 3 (after the step-2 checkpoint)
    qb = ctxmp.Queue()
    ckpt_b = str(tmp_path / "b")
    procs = _launch(str(tmp_path / "store_b"), ckpt_b, 3, "", qb)
    exitcodes = []
    for p in procs:
        p.join(timeout=120)
        exitcodes.append(p.exitcode)
    assert 17 in exitcodes, f"fault was not injected: {exitcodes}"

    # the "launcher": find the last COMPLETE manifest and restart
    steps = sorted(
        d for d in os.listdir(ckpt_b)
        if os.path.exists(layout.manifest_path(os.path.join(ckpt_b, d)))
    )
    assert steps, "no complete checkpoint survived the failure"
    last = os.path.join(ckpt_b, steps[-1])

    qc = ctxmp.Queue()
    procs = _launch(str(tmp_path / "store_c"), ckpt_b, None, last, qc)
    step_c, params_c = qc.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    assert step_a == step_c == 6
    # bit-exact continuation: same data stream (RNG state in shards),
    # same optimizer state, same params
    assert (torch.from_numpy(params_a) == torch.from_numpy(params_c)).all()
