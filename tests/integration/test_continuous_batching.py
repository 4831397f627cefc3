"""Continuous batching (scaffold.serve.ContinuousBatcher): slots at
different depths decode together; results must equal per-request
Generator.generate exactly (greedy, CPU fp32)."""

import time

import pytest
import torch

from scaffold.config import load_config
from scaffold.generate import Generator
from scaffold.models import build_model
from scaffold.serve import ContinuousBatcher


def _model(name="llama"):
    extra = (["model.n_kv_heads=2", "model.d_ff=128"] if name == "llama"
             else [])
    cfg = load_config(None, [
        f"model.name={name}", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=64", "model.vocab_size=128", "model.max_seq_len=64",
        *extra,
    ])
    torch.manual_seed(41)
    return build_model(cfg).eval()


def _ref(m, tokens, n):
    out = Generator(m).generate(
        torch.tensor([tokens], dtype=torch.int64), n)
    return out[0, len(tokens):].tolist()


@pytest.mark.parametrize("family", ["llama", "gpt2"])
def test_mixed_depth_batch_matches_generate(family):
    m = _model(family)
    cb = ContinuousBatcher(m, max_slots=4)
    try:
        prompts = [[1, 2, 3], [4, 5, 6, 7, 8, 9], [10] * 12, [11, 12]]
        nnew = [10, 4, 7, 1]
        reqs = [cb.submit(p, n) for p, n in zip(prompts, nnew)]
        for r in reqs:
            assert r.event.wait(timeout=60) and r.error is None
        for p, n, r in zip(prompts, nnew, reqs):
            assert r.result == _ref(m, p, n), (p, n)
    finally:
        cb.close()


def test_mid_flight_admission_and_slot_reuse():
    """More requests than slots, arriving while others decode: every
    result still equals the isolated rollout (no cross-slot leakage,
    clean slot recycling)."""
    m = _model()
    cb = ContinuousBatcher(m, max_slots=2)
    try:
        first = [cb.submit([1, 2, 3, 4], 16), cb.submit([5, 6], 16)]
        time.sleep(0.05)                     # let decoding start
        late = [cb.submit([7, 8, 9], 6), cb.submit([10, 11], 3),
                cb.submit([12] * 8, 5)]
        for r in first + late:
            assert r.event.wait(timeout=120) and r.error is None
        expect = [([1, 2, 3, 4], 16), ([5, 6], 16), ([7, 8, 9], 6),
                  ([10, 11], 3), ([12] * 8, 5)]
        for (p, n), r in zip(expect, first + late):
            assert r.result == _ref(m, p, n), (p, n)
        assert cb.stats["requests"] == 5
    finally:
        cb.close()


def test_bad_request_surfaces_not_fatal():
    m = _model()
    cb = ContinuousBatcher(m, max_slots=2)
    try:
        with pytest.raises(ValueError, match="exceeds"):
            cb.submit([1] * 60, 10)          # 70 > max_len 64
        ok = cb.submit([1, 2], 3)
        assert ok.event.wait(60) and ok.error is None
        assert len(ok.result) == 3
    finally:
        cb.close()


@pytest.mark.gpu
def test_continuous_batcher_gpu_deterministic():
    """GPU: runs through the HIP op path and two identical runs agree
    (greedy decode is deterministic)."""
    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=128", "model.d_ff=256",
        "model.vocab_size=512", "model.max_seq_len=128",
    ])
    torch.manual_seed(43)
    m = build_model(cfg).to("cuda", torch.bfloat16).eval()

    def run():
        cb = ContinuousBatcher(m, max_slots=4)
        try:
            reqs = [cb.submit([i + 1, i + 2, i + 3], 12) for i in range(6)]
            for r in reqs:
                assert r.event.wait(timeout=120) and r.error is None
            return [r.result for r in reqs]
        finally:
            cb.close()

    a = run()
    b = run()
    assert a == b
    assert all(len(x) == 12 for x in a)


def test_http_surface_continuous():
    from starlette.testclient import TestClient

    from scaffold.serve import create_app

    app = create_app(_model(), continuous=True, max_batch=4)
    with TestClient(app) as client:
        r = client.post("/v1/generate", json={
            "tokens": [[1, 2, 3], [4, 5, 6, 7]], "max_new_tokens": 5})
        assert r.status_code == 200, r.text
        body = r.json()
        assert len(body["tokens"]) == 2
        assert all(len(t) == 5 for t in body["tokens"])
        s = client.get("/v1/stats").json()
        assert s["requests"] == 2 and "mean_occupancy" in s


def test_sampled_requests_complete():
    """Non-greedy requests take the eager sampled path; mixing sampled
    and greedy requests in one pool must still complete correctly."""
    m = _model()
    cb = ContinuousBatcher(m, max_slots=3, seed=5)
    try:
        a = cb.submit([1, 2, 3], 8, temperature=0.9, top_k=20)
        b = cb.submit([4, 5], 6, temperature=0.7, top_p=0.9)
        c = cb.submit([6, 7, 8], 5)               # greedy alongside
        for r in (a, b, c):
            assert r.event.wait(timeout=60) and r.error is None
        assert len(a.result) == 8 and len(b.result) == 6
        assert all(0 <= t < 128 for t in a.result + b.result)
        # the greedy request is unaffected by its sampled neighbours
        assert c.result == _ref(m, [6, 7, 8], 5)
    finally:
        cb.close()


def test_eos_trim_at_harvest():
    m = _model()
    ref = _ref(m, [1, 2, 3], 10)
    eos = ref[2]                      # third generated token
    cb = ContinuousBatcher(m, max_slots=2)
    try:
        r = cb.submit([1, 2, 3], 10, eos_token_id=eos)
        assert r.event.wait(60) and r.error is None
        assert r.result == ref[:2]    # trimmed before the eos
    finally:
        cb.close()


def test_close_fails_outstanding_requests():
    """Shutdown mid-decode must fail waiters, not hang them."""
    m = _model()
    cb = ContinuousBatcher(m, max_slots=1)
    r1 = cb.submit([1, 2, 3], 60)        # long-running occupant
    r2 = cb.submit([4, 5, 6], 4)         # queued behind it
    time.sleep(0.05)
    cb.close()
    assert r1.event.wait(10) and r2.event.wait(10)
    assert (r1.result is not None) or (r1.error is not None)
    assert r2.error is not None or r2.result is not None
    with pytest.raises(RuntimeError, match="shut down"):
        cb.submit([7], 2)


def test_streaming_tokens_match_result():
    """Streamed chunks concatenate to exactly the final result, with a
    None sentinel; EOS cuts the stream."""
    import queue

    m = _model()
    ref = _ref(m, [1, 2, 3], 12)
    cb = ContinuousBatcher(m, max_slots=2)
    try:
        q = queue.Queue()
        r = cb.submit([1, 2, 3], 12, stream=q, stream_every=3)
        got = []
        while True:
            t = q.get(timeout=60)
            if t is None:
                break
            got.append(t)
        assert r.event.wait(30) and r.error is None
        assert got == r.result == ref

        # eos mid-stream: stream stops before the eos id
        eos = ref[5]
        q2 = queue.Queue()
        r2 = cb.submit([1, 2, 3], 12, eos_token_id=eos, stream=q2,
                       stream_every=2)
        got2 = []
        while True:
            t = q2.get(timeout=60)
            if t is None:
                break
            got2.append(t)
        assert r2.event.wait(30)
        assert got2 == ref[:5]
        assert r2.result == ref[:5]
    finally:
        cb.close()


def test_http_sse_stream():
    import json

    from starlette.testclient import TestClient

    from scaffold.serve import create_app

    m = _model()
    ref = _ref(m, [1, 2, 3], 8)
    app = create_app(m, continuous=True, max_batch=2)
    with TestClient(app) as client:
        with client.stream("POST", "/v1/generate", json={
                "tokens": [1, 2, 3], "max_new_tokens": 8,
                "stream": True}) as r:
            assert r.status_code == 200
            assert "text/event-stream" in r.headers["content-type"]
            toks = []
            for line in r.iter_lines():
                if not line.startswith("data: "):
                    continue
                payload = line[len("data: "):]
                if payload == "[DONE]":
                    break
                toks.append(json.loads(payload)["token"])
        assert toks == ref
        # non-continuous worker rejects streaming
    app2 = create_app(m, continuous=False)
    with TestClient(app2) as client:
        assert client.post("/v1/generate", json={
            "tokens": [1, 2, 3], "max_new_tokens": 4,
            "stream": True}).status_code == 400


def test_prefix_cache_exactness_and_hits():
    """Prefix-cache admission: repeated prompts and extended prompts
    reuse cached KV yet produce byte-identical results; LRU bounded."""
    m = _model()
    cb = ContinuousBatcher(m, max_slots=2, prefix_cache_entries=4)
    try:
        sys_prompt = [7, 8, 9, 10, 11, 12]
        # cold: full prefill
        a = cb.submit(sys_prompt, 6)
        assert a.event.wait(60) and a.error is None
        assert cb.stats.get("prefix_hits", 0) == 0
        # same prompt again: full-prompt hit
        b = cb.submit(sys_prompt, 6)
        assert b.event.wait(60) and b.error is None
        assert b.result == a.result == _ref(m, sys_prompt, 6)
        assert cb.stats["prefix_hits"] == 1
        # extended prompt: prefix hit, tail prefilled
        ext = sys_prompt + [3, 4]
        c = cb.submit(ext, 6)
        assert c.event.wait(60) and c.error is None
        assert c.result == _ref(m, ext, 6)
        assert cb.stats["prefix_hits"] == 2
        assert cb.stats["prefix_tokens_reused"] >= len(sys_prompt) * 2 - 1
        # LRU stays bounded
        for i in range(8):
            r = cb.submit([20 + i, 21 + i, 22 + i], 2)
            assert r.event.wait(60)
        assert len(cb._prefix_cache) <= 4
    finally:
        cb.close()


def test_prefix_cache_disabled():
    m = _model()
    cb = ContinuousBatcher(m, max_slots=2, prefix_cache_entries=0)
    try:
        p = [5, 6, 7]
        a = cb.submit(p, 4)
        b = cb.submit(p, 4)
        assert a.event.wait(60) and b.event.wait(60)
        assert a.result == b.result == _ref(m, p, 4)
        assert cb.stats.get("prefix_hits", 0) == 0
        assert len(cb._prefix_cache) == 0
    finally:
        cb.close()


def test_everything_on_serving_flow():
    """Compose the serving features in one pool: streaming + eos +
    prefix-cache + sampling + mixed lengths + slot reuse."""
    import queue

    m = _model()
    cb = ContinuousBatcher(m, max_slots=2, prefix_cache_entries=4, seed=3)
    try:
        sysp = [2, 3, 4, 5]
        ref = _ref(m, sysp, 14)
        eos = ref[6]
        cut = ref.index(eos)          # trim happens at FIRST occurrence
        # 1: plain greedy (cold, populates prefix cache)
        r1 = cb.submit(sysp, 14)
        # 2: streamed greedy with eos on the SAME prompt (prefix hit)
        q2 = queue.Queue()
        r2 = cb.submit(sysp, 14, eos_token_id=eos, stream=q2,
                       stream_every=2)
        # 3: sampled request on an extended prompt
        r3 = cb.submit(sysp + [9, 9], 8, temperature=0.8, top_k=16)
        # 4: another greedy, different length
        r4 = cb.submit([30, 31, 32], 5)
        got2 = []
        while True:
            t = q2.get(timeout=60)
            if t is None:
                break
            got2.append(t)
        for r in (r1, r2, r3, r4):
            assert r.event.wait(timeout=120) and r.error is None
        assert r1.result == ref
        assert r2.result == ref[:cut] == got2
        assert len(r3.result) == 8 and all(0 <= t < 128 for t in r3.result)
        assert r4.result == _ref(m, [30, 31, 32], 5)
        assert cb.stats["prefix_hits"] >= 1
        assert cb.stats["requests"] == 4
    finally:
        cb.close()


@pytest.mark.timeout(300)
def test_concurrent_submitters_stress():
    """3 submitter threads x 10 randomized greedy requests against 3
    slots: every result must equal its isolated rollout (no cross-slot
    or cross-thread contamination under real contention)."""
    import random
    import threading

    m = _model()
    cb = ContinuousBatcher(m, max_slots=3, prefix_cache_entries=4)
    results = {}
    lock = threading.Lock()

    def submitter(tid):
        rng = random.Random(tid)
        for i in range(10):
            s0 = rng.randint(1, 10)
            n = rng.randint(1, 12)
            prompt = [rng.randint(0, 127) for _ in range(s0)]
            r = cb.submit(prompt, n)
            assert r.event.wait(timeout=120) and r.error is None
            with lock:
                results[(tid, i)] = (prompt, n, r.result)

    try:
        threads = [threading.Thread(target=submitter, args=(t,))
                   for t in range(3)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=200)
            assert not t.is_alive()
    finally:
        cb.close()

    assert len(results) == 30
    for prompt, n, got in results.values():
        assert got == _ref(m, prompt, n), (prompt, n)
