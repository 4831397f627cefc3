"""GPipe pipeline equivalence (SURVEY §2.4 additive layer): a 2-stage
pipelined MLP over 4 micro-batches produces the same loss and the same
weight gradients as the unpipelined model on the full batch."""

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2
MB, NMICRO, DIN, DH, DOUT = 4, 4, 16, 32, 8


def _full_model_weights():
    g = torch.Generator().manual_seed(555)
    w1 = torch.randn(DH, DIN, generator=g) * 0.2
    w2 = torch.randn(DOUT, DH, generator=g) * 0.2
    X = torch.randn(NMICRO * MB, DIN, generator=g)
    Y = torch.randn(NMICRO * MB, DOUT, generator=g)
    return w1, w2, X, Y


class _Stage0(torch.nn.Module):
    def __init__(self, w1):
        super().__init__()
        self.lin = torch.nn.Linear(DIN, DH, bias=False)
        with torch.no_grad():
            self.lin.weight.copy_(w1)

    def forward(self, x):
        return torch.relu(self.lin(x))


class _Stage1(torch.nn.Module):
    def __init__(self, w2):
        super().__init__()
        self.lin = torch.nn.Linear(DH, DOUT, bias=False)
        with torch.no_grad():
            self.lin.weight.copy_(w2)

    def forward(self, h, y):
        return torch.nn.functional.mse_loss(self.lin(h), y)


def _worker(rank, world, store_path, q_ret):
    from scaffold.dist.pp import PipelineStage

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    w1, w2, X, Y = _full_model_weights()
    if rank == 0:
        mod = _Stage0(w1)
        stage = PipelineStage(mod, rank, world, (MB, DH))
        micro_in = [X[m * MB:(m + 1) * MB] for m in range(NMICRO)]
        stage.run_batch(micro_inputs=micro_in, n_micro=NMICRO)
        q_ret.put(("w1", mod.lin.weight.grad.numpy(), 0.0))
    else:
        mod = _Stage1(w2)
        stage = PipelineStage(mod, rank, world, (MB, DH))
        micro_lab = [Y[m * MB:(m + 1) * MB] for m in range(NMICRO)]
        loss = stage.run_batch(micro_labels=micro_lab, n_micro=NMICRO)
        q_ret.put(("w2", mod.lin.weight.grad.numpy(), loss))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_pipeline_matches_unpipelined(tmp_path):
    ctxmp = mp.get_context("spawn")
    ret = ctxmp.Queue()
    procs = [ctxmp.Process(target=_worker, args=(r, WORLD, str(tmp_path / "fs"), ret))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    grads = {}
    loss_pp = None
    for _ in range(WORLD):
        name, g, loss = ret.get(timeout=120)
        grads[name] = torch.from_numpy(g)
        if name == "w2":
            loss_pp = loss
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # unpipelined reference on the FULL batch: mean of micro-losses ==
    # per-micro mse averaged (equal micro sizes)
    w1, w2, X, Y = _full_model_weights()
    s0 = _Stage0(w1)
    s1 = _Stage1(w2)
    losses = []
    for m in range(NMICRO):
        losses.append(s1(s0(X[m * MB:(m + 1) * MB]), Y[m * MB:(m + 1) * MB]))
    loss_ref = torch.stack(losses).mean()
    loss_ref.backward()

    assert abs(loss_pp - loss_ref.item()) < 1e-6
    assert torch.allclose(grads["w1"], s0.lin.weight.grad, atol=1e-6)
    assert torch.allclose(grads["w2"], s1.lin.weight.grad, atol=1e-6)


def _worker_1f1b(rank, world, store_path, q_ret):
    from scaffold.dist.pp import PipelineStage

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    w1, w2, X, Y = _full_model_weights()
    if rank == 0:
        mod = _Stage0(w1)
        stage = PipelineStage(mod, rank, world, (MB, DH))
        micro_in = [X[m * MB:(m + 1) * MB] for m in range(NMICRO)]
        stage.run_batch_1f1b(micro_inputs=micro_in, n_micro=NMICRO)
        q_ret.put(("w1", mod.lin.weight.grad.numpy(), stage.peak_live))
    else:
        mod = _Stage1(w2)
        stage = PipelineStage(mod, rank, world, (MB, DH))
        micro_lab = [Y[m * MB:(m + 1) * MB] for m in range(NMICRO)]
        stage.run_batch_1f1b(micro_labels=micro_lab, n_micro=NMICRO)
        q_ret.put(("w2", mod.lin.weight.grad.numpy(), stage.peak_live))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_1f1b_matches_unpipelined_with_bounded_memory(tmp_path):
    """1F1B: same gradients as the unpipelined model AND at most
    world-rank activations live per stage (vs GPipe's n_micro)."""
    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_worker_1f1b,
                          args=(r, WORLD, str(tmp_path / "s1f1b"), q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(WORLD):
        name, grad, peak = q.get(timeout=120)
        got[name] = (grad, peak)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # unpipelined reference
    w1, w2, X, Y = _full_model_weights()
    s0, s1 = _Stage0(w1), _Stage1(w2)
    loss = s1(s0(X), Y)
    loss.backward()
    assert np.allclose(got["w1"][0], s0.lin.weight.grad.numpy(),
                       rtol=1e-5, atol=1e-7)
    assert np.allclose(got["w2"][0], s1.lin.weight.grad.numpy(),
                       rtol=1e-5, atol=1e-7)
    # the memory claim: stage r keeps <= world - r live, NOT n_micro
    assert got["w1"][1] <= WORLD          # rank 0: warmup+1 = 2
    assert got["w2"][1] <= 1              # last stage: 1 at a time
    assert NMICRO > WORLD                 # the bound is meaningful


def _worker_1f1b_w3(rank, world, store_path, q_ret):
    from scaffold.dist.pp import PipelineStage

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    g = torch.Generator().manual_seed(777)
    w1 = torch.randn(DH, DIN, generator=g) * 0.2
    w2 = torch.randn(DH, DH, generator=g) * 0.2
    w3 = torch.randn(DOUT, DH, generator=g) * 0.2
    X = torch.randn(NMICRO * MB, DIN, generator=g)
    Y = torch.randn(NMICRO * MB, DOUT, generator=g)

    class Mid(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.lin = torch.nn.Linear(DH, DH, bias=False)
            with torch.no_grad():
                self.lin.weight.copy_(w2)

        def forward(self, x):
            return torch.relu(self.lin(x))

    if rank == 0:
        mod = _Stage0(w1)
        stage = PipelineStage(mod, rank, world, (MB, DH))
        stage.run_batch_1f1b(
            micro_inputs=[X[m * MB:(m + 1) * MB] for m in range(NMICRO)],
            n_micro=NMICRO)
    elif rank == 1:
        mod = Mid()
        stage = PipelineStage(mod, rank, world, (MB, DH))
        stage.run_batch_1f1b(n_micro=NMICRO)
    else:
        mod = _Stage1(w3)
        stage = PipelineStage(mod, rank, world, (MB, DH))
        stage.run_batch_1f1b(
            micro_labels=[Y[m * MB:(m + 1) * MB] for m in range(NMICRO)],
            n_micro=NMICRO)
    q_ret.put((rank, mod.lin.weight.grad.numpy(), stage.peak_live))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_1f1b_three_stages(tmp_path):
    """Deeper pipeline: the steady-state interleave crosses two
    boundaries at once; grads must still match the unpipelined model
    and the per-stage activation bound must hold."""
    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_worker_1f1b_w3,
                          args=(r, 3, str(tmp_path / "s3"), q))
             for r in range(3)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(3):
        rank, grad, peak = q.get(timeout=120)
        got[rank] = (grad, peak)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    g = torch.Generator().manual_seed(777)
    w1 = torch.randn(DH, DIN, generator=g) * 0.2
    w2 = torch.randn(DH, DH, generator=g) * 0.2
    w3 = torch.randn(DOUT, DH, generator=g) * 0.2
    X = torch.randn(NMICRO * MB, DIN, generator=g)
    Y = torch.randn(NMICRO * MB, DOUT, generator=g)
    l1 = torch.nn.Linear(DIN, DH, bias=False)
    l2 = torch.nn.Linear(DH, DH, bias=False)
    l3 = torch.nn.Linear(DH, DOUT, bias=False)
    with torch.no_grad():
        l1.weight.copy_(w1)
        l2.weight.copy_(w2)
        l3.weight.copy_(w3)
    loss = torch.nn.functional.mse_loss(
        l3(torch.relu(l2(torch.relu(l1(X))))), Y)
    loss.backward()
    for rank, ref in ((0, l1), (1, l2), (2, l3)):
        assert np.allclose(got[rank][0], ref.weight.grad.numpy(),
                           rtol=1e-5, atol=1e-7), rank
    # bound: stage r holds at most world - r live activations
    assert got[0][1] <= 3 and got[1][1] <= 2 and got[2][1] <= 1
