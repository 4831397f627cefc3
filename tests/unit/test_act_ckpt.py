"""Activation checkpointing (model.act_ckpt): recomputed backward is
bitwise-identical to the stored-activation backward (no dropout in
either LM family), and eval/decode paths are untouched."""

import copy

import pytest
import torch

from scaffold.config import load_config
from scaffold.models import build_model


def _grads(name, act_ckpt, extra=()):
    ov = [
        f"model.name={name}", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=64", "model.vocab_size=128", "model.max_seq_len=32",
        f"model.act_ckpt={'true' if act_ckpt else 'false'}", *extra,
    ]
    cfg = load_config(None, ov)
    torch.manual_seed(5)
    m = build_model(cfg)
    torch.manual_seed(6)
    tok = torch.randint(0, 128, (2, 32))
    tgt = torch.randint(0, 128, (2, 32))
    loss = m(tok, tgt)
    loss.backward()
    return loss.detach(), [p.grad.clone() for p in m.parameters()]


@pytest.mark.parametrize("name,extra", [
    ("gpt2", ()),
    ("llama", ("model.n_kv_heads=2", "model.d_ff=128")),
])
def test_act_ckpt_bitwise_matches(name, extra):
    l0, g0 = _grads(name, False, extra)
    l1, g1 = _grads(name, True, extra)
    assert torch.equal(l0, l1)
    assert len(g0) == len(g1)
    for a, b in zip(g0, g1):
        assert torch.equal(a, b)


def test_act_ckpt_off_in_eval():
    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=1", "model.n_heads=2",
        "model.d_model=32", "model.vocab_size=64", "model.max_seq_len=16",
        "model.act_ckpt=true",
    ])
    m = build_model(cfg).eval()
    tok = torch.randint(0, 64, (1, 16))
    with torch.no_grad():
        loss = m(tok, tok)           # no checkpoint machinery under no_grad
    assert torch.isfinite(loss)


def _ddp_worker(rank, world, store_path, act_ckpt, q):
    import torch.distributed as dist

    from scaffold.runtime import DistContext
    from scaffold.trainer import Trainer

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=64", "model.vocab_size=128", "model.max_seq_len=32",
        f"model.act_ckpt={'true' if act_ckpt else 'false'}",
        "data.kind=synthetic_tokens", "data.batch_size=2", "data.seq_len=32",
        "dtype=fp32", "device=cpu", "log.interval=1000",
        "dist.bucket_cap_mb=0.125",
    ])
    ctx = DistContext(rank=rank, world_size=world, local_rank=rank,
                      backend="gloo", device=torch.device("cpu"),
                      initialized=True)
    t = Trainer(cfg, ctx=ctx)
    g = torch.Generator().manual_seed(13 + rank)
    tok = torch.randint(0, 128, (2, 32), generator=g)
    t.ddp.zero_grad()
    loss = t.model(tok, tok)
    loss.backward()
    t.ddp.finish()
    grads = torch.cat([p.grad.flatten() for p in t.raw_model.parameters()])
    if rank == 0:
        q.put(grads.numpy())
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_act_ckpt_under_bucketed_ddp(tmp_path):
    """Recompute-in-backward must fire each param's bucket hook exactly
    once: all-reduced grads identical with and without act_ckpt."""
    import numpy as np
    import torch.multiprocessing as mp

    outs = []
    for ac in (False, True):
        ctxm = mp.get_context("spawn")
        q = ctxm.Queue()
        store = str(tmp_path / f"store_{ac}")
        procs = [ctxm.Process(target=_ddp_worker, args=(r, 2, store, ac, q))
                 for r in range(2)]
        for p in procs:
            p.start()
        outs.append(q.get(timeout=150))
        for p in procs:
            p.join(timeout=60)
            assert p.exitcode == 0
    assert np.array_equal(outs[0], outs[1])


@pytest.mark.gpu
def test_act_ckpt_gpu_peak_memory_drops():
    """On MI355X the point of act_ckpt is activation memory: peak
    allocation for fwd+bwd must drop materially with recompute on."""
    def peak(ac):
        cfg = load_config(None, [
            "model.name=llama", "model.n_layers=8", "model.n_heads=8",
            "model.n_kv_heads=4", "model.d_model=512", "model.d_ff=1024",
            "model.vocab_size=1024", "model.max_seq_len=1024",
            f"model.act_ckpt={'true' if ac else 'false'}",
        ])
        torch.manual_seed(2)
        m = build_model(cfg).to("cuda", torch.bfloat16)
        tok = torch.randint(0, 1024, (8, 1024), device="cuda")
        torch.cuda.synchronize()
        torch.cuda.reset_peak_memory_stats()
        base = torch.cuda.memory_allocated()
        loss = m(tok, tok)
        loss.backward()
        torch.cuda.synchronize()
        pk = torch.cuda.max_memory_allocated() - base
        del m, loss
        torch.cuda.empty_cache()
        return pk

    p_plain = peak(False)
    p_ckpt = peak(True)
    # stored activations dominate at S=1024: expect a large cut
    assert p_ckpt < 0.7 * p_plain, (p_plain, p_ckpt)
