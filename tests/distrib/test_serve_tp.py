"""TP-distributed serving (scaffold/serve_tp.py): rank 0 batcher +
follower replaying the command stream on its weight shard must equal
the unsharded rollout exactly, through admissions, slot reuse and
shutdown."""

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from scaffold.config import load_config

OVER = [
    "model.name=llama", "model.n_layers=2", "model.n_heads=4",
    "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
    "model.vocab_size=128", "model.max_seq_len=64",
]
PROMPTS = [[1, 2, 3], [4, 5, 6, 7, 8], [9] * 7, [10, 11]]
NNEW = [12, 5, 8, 3]


def _worker(rank, world, store_path, q):
    from scaffold.dist.tp import shard_llama_tp
    from scaffold.models import build_model
    from scaffold.serve_tp import TpContinuousBatcher, run_follower

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    cfg = load_config(None, OVER)
    torch.manual_seed(111)
    m = build_model(cfg).eval()
    shard_llama_tp(m, rank, world)
    if rank == 0:
        cb = TpContinuousBatcher(m, max_slots=2)
        try:
            reqs = [cb.submit(p, n) for p, n in zip(PROMPTS, NNEW)]
            for r in reqs:
                assert r.event.wait(timeout=200) and r.error is None
            results = [r.result for r in reqs]
            with pytest.raises(ValueError, match="greedy-only"):
                cb.submit([1, 2], 4, temperature=0.7)
        finally:
            cb.close()          # broadcasts STOP from the engine thread
        q.put(results)
    else:
        run_follower(m, max_slots=2)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_tp_serving_matches_unsharded(tmp_path):
    from scaffold.generate import Generator
    from scaffold.models import build_model

    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_worker,
                          args=(r, 2, str(tmp_path / "stp"), q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = q.get(timeout=250)
    for p in procs:
        p.join(timeout=90)
        assert p.exitcode == 0

    cfg = load_config(None, OVER)
    torch.manual_seed(111)
    full = build_model(cfg).eval()
    g = Generator(full)
    for prompt, n, got in zip(PROMPTS, NNEW, results):
        ref = g.generate(torch.tensor([prompt], dtype=torch.int64), n)
        assert got == ref[0, len(prompt):].tolist(), (prompt, n)


def test_create_app_accepts_injected_worker():
    """The endpoints must serve the INJECTED worker (regression for the
    closed-over-worker bug in the first TP CLI cut)."""
    from starlette.testclient import TestClient

    from scaffold.models import build_model
    from scaffold.serve import ContinuousBatcher, create_app

    cfg = load_config(None, OVER)
    torch.manual_seed(5)
    m = build_model(cfg).eval()
    w = ContinuousBatcher(m, max_slots=2)
    app = create_app(m, worker=w)
    assert app.state.worker is w
    with TestClient(app) as client:
        r = client.post("/v1/generate", json={"tokens": [1, 2, 3],
                                              "max_new_tokens": 4})
        assert r.status_code == 200
        assert w.stats["requests"] == 1   # THE injected worker served it


@pytest.mark.timeout(300)
def test_tp_serving_cli_end_to_end(tmp_path):
    """The real CLI under a 2-rank env rendezvous: rank 0 uvicorn
    answers over TCP, SIGTERM drains cleanly and the STOP broadcast
    releases the follower (its clean exit proves the protocol)."""
    import os
    import signal
    import socket
    import subprocess
    import sys
    import time

    import httpx

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        http_port = s.getsockname()[1]
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        pg_port = s.getsockname()[1]

    repo = os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update({"RANK": str(rank), "WORLD_SIZE": "2",
                    "MASTER_ADDR": "127.0.0.1",
                    "MASTER_PORT": str(pg_port)})
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "scaffold.serve_tp", *OVER,
             "--port", str(http_port), "--max-slots", "2"],
            cwd=repo, env=env))
    try:
        url = f"http://127.0.0.1:{http_port}"
        deadline = time.time() + 90
        while time.time() < deadline:
            try:
                if httpx.get(url + "/healthz", timeout=2).status_code == 200:
                    break
            except Exception:
                time.sleep(0.3)
        else:
            raise AssertionError("TP server never became healthy")
        r = httpx.post(url + "/v1/generate",
                       json={"tokens": [1, 2, 3], "max_new_tokens": 6},
                       timeout=60)
        assert r.status_code == 200, r.text
        got = r.json()["tokens"][0]
        assert len(got) == 6

        from scaffold.generate import Generator
        from scaffold.models import build_model

        cfg = load_config(None, OVER)
        torch.manual_seed(cfg.seed)
        full = build_model(cfg).eval()
        ref = Generator(full).generate(
            torch.tensor([[1, 2, 3]], dtype=torch.int64), 6)
        assert got == ref[0, 3:].tolist()
    finally:
        procs[0].send_signal(signal.SIGTERM)
        rc0 = procs[0].wait(timeout=60)
        rc1 = procs[1].wait(timeout=60)   # released by the STOP broadcast
    assert rc1 == 0, "follower did not exit cleanly (STOP never arrived)"
    assert rc0 in (0, -signal.SIGTERM)


def _stream_worker(rank, world, store_path, q):
    import queue as pyqueue

    from scaffold.dist.tp import shard_llama_tp
    from scaffold.models import build_model
    from scaffold.serve_tp import TpContinuousBatcher, run_follower

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    cfg = load_config(None, OVER)
    torch.manual_seed(111)
    m = build_model(cfg).eval()
    shard_llama_tp(m, rank, world)
    if rank == 0:
        cb = TpContinuousBatcher(m, max_slots=2)
        try:
            sq = pyqueue.Queue()
            r1 = cb.submit([1, 2, 3], 10, stream=sq, stream_every=3)
            streamed = []
            while True:
                t = sq.get(timeout=120)
                if t is None:
                    break
                streamed.append(t)
            assert r1.event.wait(60) and r1.error is None
            # prefix-cache reuse on a repeated prompt, through TP
            r2 = cb.submit([1, 2, 3], 10)
            assert r2.event.wait(120) and r2.error is None
            q.put((streamed, r1.result, r2.result,
                   cb.stats.get("prefix_hits", 0)))
        finally:
            cb.close()
    else:
        run_follower(m, max_slots=2)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_tp_serving_streaming_and_prefix_cache(tmp_path):
    from scaffold.generate import Generator
    from scaffold.models import build_model

    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_stream_worker,
                          args=(r, 2, str(tmp_path / "stps"), q))
             for r in range(2)]
    for p in procs:
        p.start()
    streamed, r1, r2, hits = q.get(timeout=250)
    for p in procs:
        p.join(timeout=90)
        assert p.exitcode == 0

    cfg = load_config(None, OVER)
    torch.manual_seed(111)
    ref = Generator(build_model(cfg).eval()).generate(
        torch.tensor([[1, 2, 3]], dtype=torch.int64), 10)[0, 3:].tolist()
    assert streamed == r1 == r2 == ref
    assert hits >= 1


def _ep_serve_worker(rank, world, store_path, q):
    import torch.distributed as dist

    from scaffold.models import build_model
    from scaffold.serve_tp import TpContinuousBatcher, run_follower

    store = dist.FileStore(store_path, world)
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world)
    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=2",
        "model.n_kv_heads=1", "model.d_model=32", "model.d_ff=48",
        "model.vocab_size=64", "model.max_seq_len=48",
        "model.n_experts=4", "model.moe_top_k=2",
    ])
    ep_group = dist.new_group(list(range(world)))
    cfg._ep_ctx = (world, rank, ep_group)
    torch.manual_seed(171)
    m = build_model(cfg).eval()
    # replicated backbone must agree across ranks (rank-seeded RNG only
    # diverges for rank-local experts, which are per-expert seeded) —
    # broadcast rank 0's non-expert params like the trainer wrap does
    with torch.no_grad():
        for p in m.parameters():
            if not getattr(p, "_no_ddp", False):
                dist.broadcast(p.data, src=0)
    if rank == 0:
        cb = TpContinuousBatcher(m, max_slots=2)
        try:
            reqs = [cb.submit([1, 2, 3], 10), cb.submit([4, 5, 6, 7], 6)]
            for r in reqs:
                assert r.event.wait(timeout=200) and r.error is None
            q.put([r.result for r in reqs])
        finally:
            cb.close()
    else:
        run_follower(m, max_slots=2)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ep_sharded_moe_serving_matches_unsharded(tmp_path):
    """Distributed MoE serving: the SAME command-replay protocol serves
    an expert-sharded model — routed-token a2a fires inside each decode
    step on every rank, results equal the single-process local-expert
    rollout (Mixtral-8x7B-class serving shape)."""
    from scaffold.generate import Generator
    from scaffold.models import build_model

    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_ep_serve_worker,
                          args=(r, 2, str(tmp_path / "eps"), q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = q.get(timeout=250)
    for p in procs:
        p.join(timeout=90)
        assert p.exitcode == 0

    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=2",
        "model.n_kv_heads=1", "model.d_model=32", "model.d_ff=48",
        "model.vocab_size=64", "model.max_seq_len=48",
        "model.n_experts=4", "model.moe_top_k=2",
    ])
    torch.manual_seed(171)
    full = build_model(cfg).eval()
    g = Generator(full)
    for (prompt, n), got in zip([([1, 2, 3], 10), ([4, 5, 6, 7], 6)],
                                results):
        ref = g.generate(torch.tensor([prompt], dtype=torch.int64), n)
        assert got == ref[0, len(prompt):].tolist(), (prompt, n)
