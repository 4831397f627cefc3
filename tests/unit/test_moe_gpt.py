"""MoE GPT-2 (model.n_experts > 0): routed-FFN blocks train, every
expert's grad is defined even when unrouted (bucketed-DDP contract),
and ws=2 DDP training runs to completion."""

import pytest
import torch

from scaffold.config import load_config


def _cfg(**kw):
    over = [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=2",
        "model.d_model=32", "model.vocab_size=64", "model.max_seq_len=16",
        "model.n_experts=4", "data.kind=synthetic_tokens",
        "data.batch_size=4", "data.seq_len=16", "data.active_vocab=8",
        "device=cpu", "dtype=fp32", "log.interval=1000", "optim.lr=0.01",
    ]
    over += [f"{k}={v}" for k, v in kw.items()]
    return load_config(None, over)


def test_moe_gpt_trains_and_loss_drops():
    from scaffold.trainer import Trainer

    t = Trainer(_cfg(steps=30))
    it = iter(t.data)
    losses = []
    for _ in range(30):
        losses.append(t._train_step(it, 1))
    assert all(torch.isfinite(torch.tensor(losses)))
    assert min(losses[-5:]) < losses[0] - 0.5, (losses[0], losses[-5:])


def test_unrouted_expert_grad_defined():
    """Force all tokens onto expert 0: the unrouted experts' grads must
    still exist (exact zeros) so a bucketed reducer never stalls."""
    from scaffold.dist.ep import ExpertParallelMoE

    torch.manual_seed(0)
    moe = ExpertParallelMoE(16, 32, 4, world=1, rank=0, dispatch="dense")
    with torch.no_grad():
        moe.gate.weight.zero_()
        moe.gate.weight[0].fill_(1.0)     # logit_0 = sum(x) > 0 for x > 0
    x = torch.rand(8, 16) + 0.1          # strictly positive inputs
    moe(x).sum().backward()
    assert moe.w_up[0].grad is not None and moe.w_up[0].grad.abs().sum() > 0
    for i in (1, 2, 3):
        for p in (moe.w_up[i], moe.w_down[i]):
            assert p.grad is not None, i
            assert torch.all(p.grad == 0), i


def _ddp_worker(rank, world, store_path, q):
    import torch.distributed as dist

    from scaffold.runtime import DistContext
    from scaffold.trainer import Trainer

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    ctx = DistContext(rank=rank, world_size=world, local_rank=rank,
                      backend="gloo", device=torch.device("cpu"),
                      initialized=True)
    t = Trainer(_cfg(), ctx=ctx)
    it = iter(t.data)
    losses = [t._train_step(it, 1) for _ in range(5)]
    if rank == 0:
        q.put(losses)
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_moe_under_ddp_ws2(tmp_path):
    import torch.multiprocessing as mp

    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_ddp_worker,
                          args=(r, 2, str(tmp_path / "store"), q))
             for r in range(2)]
    for p in procs:
        p.start()
    losses = q.get(timeout=150)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert all(torch.isfinite(torch.tensor(losses)))
