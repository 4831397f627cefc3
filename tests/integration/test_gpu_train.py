"""End-to-end GPU training smoke per model family (@gpu): a few steps
through the full path (HIP ops + BucketedDDP ws=1 + fused AdamW), loss
finite and decreasing for the small configs."""

import pytest
import torch

from scaffold.config import load_config
from scaffold.trainer import Trainer

pytestmark = pytest.mark.gpu


def _fit_steps(cfg, n):
    t = Trainer(cfg)
    it = iter(t.data)
    losses = []
    for _ in range(n):
        losses.append(t._train_step(it, 1))
    torch.cuda.synchronize()
    return losses


def test_resnet50_gpu_steps():
    cfg = load_config(None, [
        "model.name=resnet50", "data.kind=synthetic_image",
        "data.batch_size=8", "dtype=bf16", "log.interval=1000",
    ])
    losses = _fit_steps(cfg, 3)
    assert all(torch.isfinite(torch.tensor(losses))), losses


def test_gpt2_tiny_gpu_loss_decreases():
    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=256", "model.vocab_size=512", "model.max_seq_len=128",
        "data.kind=synthetic_tokens", "data.batch_size=4", "data.seq_len=128",
        "data.active_vocab=16",
        "optim.lr=0.003", "dtype=bf16", "log.interval=1000",
    ])
    losses = _fit_steps(cfg, 25)
    assert all(torch.isfinite(torch.tensor(losses))), losses
    assert min(losses[-3:]) < losses[0] - 1.0, losses


def test_llama_tiny_gpu_loss_decreases():
    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=256", "model.d_ff=512",
        "model.vocab_size=512", "model.max_seq_len=128",
        "data.kind=synthetic_tokens", "data.batch_size=4", "data.seq_len=128",
        "data.active_vocab=16",
        "optim.lr=0.003", "dtype=bf16", "log.interval=1000",
    ])
    losses = _fit_steps(cfg, 25)
    assert all(torch.isfinite(torch.tensor(losses))), losses
    assert min(losses[-3:]) < losses[0] - 1.0, losses


def test_ckpt_roundtrip_gpu(tmp_path):
    import copy

    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=256", "model.vocab_size=512", "model.max_seq_len=128",
        "data.kind=synthetic_tokens", "data.batch_size=4", "data.seq_len=128",
        "dtype=bf16", "log.interval=1000",
    ])
    cfg.ckpt.dir = str(tmp_path)
    t = Trainer(cfg)
    it = iter(t.data)
    for _ in range(3):
        t._train_step(it, 1)
        t.step += 1
    ck = t.save_checkpoint()

    cfg2 = copy.deepcopy(cfg)
    cfg2.ckpt.resume = ck
    cfg2.seed = 7
    t2 = Trainer(cfg2)
    assert t2.step == 3
    for a, b in zip(t.raw_model.parameters(), t2.raw_model.parameters()):
        assert torch.equal(a.detach(), b.detach())


def test_rccl_direct_bucket_ws1(tmp_path):
    """ws=1 RCCL: the direct C++ bucket path (comm stream + join) end to
    end — exercises ncclCommInitRank/AllReduce minus real p2p (SURVEY §4
    'single-GPU multi-rank')."""
    import torch.distributed as dist
    from scaffold.dist.comm import RcclBucketComm

    store = dist.FileStore(str(tmp_path / "store"), 1)
    dist.init_process_group("nccl", store=store, rank=0, world_size=1)
    try:
        cfg = load_config(None, [
            "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
            "model.d_model=256", "model.vocab_size=512", "model.max_seq_len=128",
            "data.kind=synthetic_tokens", "data.batch_size=4", "data.seq_len=128",
            "data.active_vocab=16", "dist.comm=rccl",
            "optim.lr=0.003", "dtype=bf16", "log.interval=1000",
        ])
        t = Trainer(cfg)
        assert isinstance(t.ddp.comm, RcclBucketComm), type(t.ddp.comm)
        it = iter(t.data)
        losses = [t._train_step(it, 1) for _ in range(10)]
        torch.cuda.synchronize()
        assert all(torch.isfinite(torch.tensor(losses))), losses
        assert min(losses[-3:]) < losses[0], losses
        # per-bucket collective time was measured on the comm stream
        assert t.ddp.comm.comm_time_ms() > 0.0
    finally:
        dist.destroy_process_group()


def test_fused_adamw_cache_invalidation():
    """The fused step caches device grad POINTERS; zero_grad(set_to_none)
    re-materializes grads, so the cache must rebuild (stale
    pointers silently updated from freed memory)."""
    from scaffold.optim import MasterAdamW

    torch.manual_seed(0)
    lin_gpu = torch.nn.Linear(64, 64).to("cuda", torch.bfloat16)
    lin_cpu = torch.nn.Linear(64, 64)
    with torch.no_grad():
        lin_cpu.weight.copy_(lin_gpu.weight.float())
        lin_cpu.bias.copy_(lin_gpu.bias.float())
    og = MasterAdamW(lin_gpu.parameters(), lr=1e-2)
    oc = MasterAdamW(lin_cpu.parameters(), lr=1e-2)

    for it in range(3):
        g_w = torch.randn(64, 64)
        g_b = torch.randn(64)
        # set_to_none between steps: grad tensors get NEW storage each time
        for p in lin_gpu.parameters():
            p.grad = None
        for p in lin_cpu.parameters():
            p.grad = None
        lin_gpu.weight.grad = g_w.to("cuda", torch.bfloat16)
        lin_gpu.bias.grad = g_b.to("cuda", torch.bfloat16)
        lin_cpu.weight.grad = g_w.to(torch.bfloat16).float()
        lin_cpu.bias.grad = g_b.to(torch.bfloat16).float()
        og.step()
        oc.step()
    torch.cuda.synchronize()
    for mg, mc in zip(og.master, oc.master):
        assert torch.allclose(mg.cpu(), mc, rtol=2e-3, atol=2e-3), (
            (mg.cpu() - mc).abs().max().item())


def test_hip_graph_step_replay():
    """hipGraph-captured train step: loss decreases across replays and
    the device step counter keeps AdamW bias corrections exact."""
    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=256", "model.vocab_size=512", "model.max_seq_len=128",
        "data.kind=synthetic_tokens", "data.batch_size=4", "data.seq_len=128",
        "data.active_vocab=16", "hip_graph=true", "steps=25",
        "optim.lr=0.003", "dtype=bf16", "log.interval=1000",
    ])
    t = Trainer(cfg)
    t.fit()
    torch.cuda.synchronize()
    if hasattr(t, "_graph"):
        # graph path active: step counters advanced under replay
        assert int(t.optimizer.step_dev.item()) == t.optimizer.step_count
    # loss decreased over the run either way (capture may legitimately
    # fall back to eager if RNG capture is unsupported)
    l0 = t.model(*next(iter(t.data))).float().item()
    assert torch.isfinite(torch.tensor(l0))
    assert l0 < 5.0, l0  # ln(16)=2.77 start; must have learned something


def test_generate_graphed_gpu_matches():
    """hipGraph-replayed decode == plain greedy decode on GPU."""
    from scaffold.generate import Generator
    from scaffold.models import build_model

    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=128", "model.d_ff=256",
        "model.vocab_size=512", "model.max_seq_len=128",
    ])
    torch.manual_seed(9)
    m = build_model(cfg).to("cuda", torch.bfloat16).eval()
    g = Generator(m)
    prompt = torch.randint(0, 512, (2, 16), device="cuda")
    a = g.generate(prompt, max_new_tokens=24)
    b = g.generate_graphed(prompt, max_new_tokens=24)
    torch.cuda.synchronize()
    assert torch.equal(a, b)


def test_moe_gpt_gpu_trains():
    """MoE FFN blocks through the HIP op path: loss decreases."""
    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=256", "model.vocab_size=512", "model.max_seq_len=128",
        "model.n_experts=4", "data.kind=synthetic_tokens",
        "data.batch_size=4", "data.seq_len=128", "data.active_vocab=16",
        "optim.lr=0.003", "dtype=bf16", "log.interval=1000",
    ])
    losses = _fit_steps(cfg, 25)
    assert all(torch.isfinite(torch.tensor(losses))), losses
    assert min(losses[-3:]) < losses[0] - 1.0, losses
