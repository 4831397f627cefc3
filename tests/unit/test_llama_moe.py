"""Llama-MoE (Mixtral-class): SwiGLU experts wired into the Llama
block — trains, decodes exactly, and shards across ranks (dist.ep_moe)
with the same machinery as the GPT-2 MoE."""

import pytest
import torch

from scaffold.config import load_config


def _cfg(**kw):
    over = [
        "model.name=llama", "model.n_layers=2", "model.n_heads=2",
        "model.n_kv_heads=1", "model.d_model=32", "model.d_ff=48",
        "model.vocab_size=64", "model.max_seq_len=48",
        "model.n_experts=4", "model.moe_top_k=2",
        "data.kind=synthetic_tokens", "data.batch_size=4",
        "data.seq_len=16", "data.active_vocab=8",
        "device=cpu", "dtype=fp32", "log.interval=1000", "optim.lr=0.02",
    ]
    over += [f"{k}={v}" for k, v in kw.items()]
    return load_config(None, over)


def test_llama_moe_trains_and_decodes():
    from scaffold.generate import Generator, KVCache
    from scaffold.trainer import Trainer

    t = Trainer(_cfg(steps=30))
    moe = t.raw_model.blocks[0].moe
    assert moe.act == "swiglu" and moe.top_k == 2
    assert moe.w_up[0].shape == (2 * 48, 32)   # packed gate|up
    it = iter(t.data)
    losses = [t._train_step(it, 1) for _ in range(30)]
    assert all(torch.isfinite(torch.tensor(losses)))
    assert min(losses[-5:]) < losses[0] - 0.4, (losses[0], losses[-5:])

    m = t.raw_model.eval()
    g = Generator(m)
    prompt = torch.randint(0, 64, (1, 5))
    out = g.generate(prompt, 8)
    toks = prompt.clone()
    with torch.no_grad():
        for _ in range(8):
            cache = KVCache(2, 1, 1, toks.shape[1], 16, toks.device,
                            torch.float32)
            logits = g._forward_tokens(toks, cache, 0)
            toks = torch.cat([toks, logits.argmax(-1, keepdim=True)], dim=1)
    assert torch.equal(out, toks)


def _ep_worker(rank, world, store_path, ckpt_dir, resume, steps, q):
    import torch.distributed as dist

    from scaffold.runtime import DistContext
    from scaffold.trainer import Trainer

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    ctx = DistContext(rank=rank, world_size=world, local_rank=rank,
                      backend="gloo", device=torch.device("cpu"),
                      initialized=True)
    over = {"dist.ep_moe": "true", "steps": steps}
    if ckpt_dir:
        over.update({"ckpt.dir": ckpt_dir, "ckpt.save_final": "true"})
    if resume:
        over["ckpt.resume"] = resume
    t = Trainer(_cfg(**over), ctx=ctx)
    moe = t.raw_model.blocks[0].moe
    assert moe.world == world and moe.act == "swiglu"
    t.fit()
    params = torch.cat([p.detach().flatten().float()
                        for _, p in sorted(t.raw_model.named_parameters())])
    if rank == 0:
        q.put(params.numpy())
    dist.barrier()
    dist.destroy_process_group()


def _run(tmp_path, tag, steps, ckpt_dir="", resume=""):
    import torch.multiprocessing as mp

    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_ep_worker,
                          args=(r, 2, str(tmp_path / tag), ckpt_dir,
                                resume, steps, q))
             for r in range(2)]
    for p in procs:
        p.start()
    out = q.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    return out


@pytest.mark.timeout(600)
def test_llama_moe_ep_resume_bitexact(tmp_path):
    import numpy as np

    from scaffold.ckpt import layout

    pa = _run(tmp_path, "a", 6, ckpt_dir=str(tmp_path / "ca"))
    _run(tmp_path, "b1", 4, ckpt_dir=str(tmp_path / "cb"))
    mid = layout.latest_checkpoint(str(tmp_path / "cb"))
    pb = _run(tmp_path, "b2", 6, ckpt_dir=str(tmp_path / "cb2"),
              resume=mid)
    assert np.array_equal(pa, pb)


@pytest.mark.timeout(600)
def test_llama_moe_reshard_to_single_process(tmp_path):
    """EP ws=2 Llama-MoE checkpoint loads into a ws=1 local-expert
    model (global-id name fallback), optimizer state included."""
    from scaffold.ckpt import layout
    from scaffold.trainer import Trainer

    _run(tmp_path, "r", 4, ckpt_dir=str(tmp_path / "cr"))
    ck = layout.latest_checkpoint(str(tmp_path / "cr"))
    t = Trainer(_cfg(**{"ckpt.resume": ck, "steps": 4}))
    assert t.step == 4
    assert t.raw_model.blocks[0].moe.world == 1
    assert any(m.abs().sum() > 0 for m in t.optimizer.exp_avg)
    # every global expert materialized locally
    import safetensors.torch as st
    import pathlib
    shards = {}
    for f in sorted(pathlib.Path(ck).iterdir()):
        if f.name.startswith("shard_"):
            shards.update(st.load_file(str(f)))
    for gid in range(4):
        saved = shards[f"model.blocks.0.moe.w_up.g{gid}"]
        assert torch.equal(
            t.raw_model.blocks[0].moe.w_up[gid].detach(), saved)


def _ep_act_worker(rank, world, store_path, q):
    import torch.distributed as dist

    from scaffold.runtime import DistContext
    from scaffold.trainer import Trainer

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    ctx = DistContext(rank=rank, world_size=world, local_rank=rank,
                      backend="gloo", device=torch.device("cpu"),
                      initialized=True)
    t = Trainer(_cfg(**{"dist.ep_moe": "true", "model.act_ckpt": "true",
                        "steps": 4}), ctx=ctx)
    t.fit()
    if rank == 0:
        q.put(t.step)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_llama_moe_ep_with_act_ckpt(tmp_path):
    import torch.multiprocessing as mp

    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_ep_act_worker,
                          args=(r, 2, str(tmp_path / "epa"), q))
             for r in range(2)]
    for p in procs:
        p.start()
    assert q.get(timeout=200) == 4
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
