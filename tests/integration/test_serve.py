"""Serving layer (scaffold/serve.py): dynamic batching correctness and
the HTTP surface, on a tiny CPU llama (fp32 eager path)."""

import threading

import pytest
import torch

from scaffold.config import load_config
from scaffold.generate import Generator
from scaffold.models import build_model
from scaffold.serve import InferenceWorker, create_app


def _tiny_model():
    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
        "model.vocab_size=128", "model.max_seq_len=64",
    ])
    torch.manual_seed(3)
    return build_model(cfg).eval()


def test_worker_batches_same_shape_requests():
    m = _tiny_model()
    w = InferenceWorker(m, max_batch=8, window_s=0.05)
    try:
        prompts = [[1, 2, 3, 4], [5, 6, 7, 8], [9, 10, 11, 12]]
        reqs = [w.submit(p, 8) for p in prompts]
        for r in reqs:
            assert r.event.wait(timeout=30)
            assert r.error is None
            assert len(r.result) == 8
        # same (len, temp) key arriving within the window -> ONE batch
        assert w.stats["requests"] == 3
        assert w.stats["batches"] == 1
        # batched output == unbatched reference decode (greedy)
        ref = Generator(m).generate(
            torch.tensor(prompts, dtype=torch.int64), 8)
        for i, r in enumerate(reqs):
            assert r.result == ref[i, 4:].tolist()
    finally:
        w.close()


def test_worker_groups_by_prompt_length():
    m = _tiny_model()
    w = InferenceWorker(m, max_batch=8, window_s=0.05)
    try:
        a = w.submit([1, 2, 3], 4)
        b = w.submit([4, 5, 6, 7, 8], 4)   # different length: own batch
        c = w.submit([7, 8, 9], 4)
        for r in (a, b, c):
            assert r.event.wait(timeout=30) and r.error is None
        assert w.stats["requests"] == 3
        assert w.stats["batches"] == 2
    finally:
        w.close()


def test_worker_mixed_max_new_tokens_sliced_per_request():
    m = _tiny_model()
    w = InferenceWorker(m, max_batch=8, window_s=0.05)
    try:
        a = w.submit([1, 2, 3, 4], 4)
        b = w.submit([1, 2, 3, 4], 9)
        assert a.event.wait(30) and b.event.wait(30)
        assert len(a.result) == 4 and len(b.result) == 9
        # the shared batch ran to max(4, 9); a's slice must equal the
        # first 4 of the same greedy rollout
        assert b.result[:4] == a.result
    finally:
        w.close()


def test_worker_error_surfaces_not_fatal():
    m = _tiny_model()
    w = InferenceWorker(m, max_batch=4, window_s=0.0)
    try:
        # context overflow is rejected synchronously at submit
        with pytest.raises(ValueError, match="exceeds"):
            w.submit([1, 2], 200)
        # a RUNTIME failure (out-of-vocab id -> embedding index error)
        # surfaces on the request without killing the worker loop
        bad = w.submit([500, 501], 4)
        assert bad.event.wait(30)
        assert bad.error is not None
        ok = w.submit([1, 2, 3], 4)   # loop survived
        assert ok.event.wait(30) and ok.error is None
    finally:
        w.close()


def test_http_surface():
    from starlette.testclient import TestClient

    app = create_app(_tiny_model(), window_s=0.0)
    with TestClient(app) as client:
        assert client.get("/healthz").json() == {"ok": True}

        r = client.post("/v1/generate", json={
            "tokens": [1, 2, 3, 4], "max_new_tokens": 6})
        assert r.status_code == 200, r.text
        body = r.json()
        assert len(body["tokens"]) == 1 and len(body["tokens"][0]) == 6
        assert all(isinstance(t, int) for t in body["tokens"][0])

        # multi-prompt body batches server-side
        r2 = client.post("/v1/generate", json={
            "tokens": [[1, 2, 3, 4], [5, 6, 7, 8]], "max_new_tokens": 5})
        assert r2.status_code == 200
        assert len(r2.json()["tokens"]) == 2

        assert client.post("/v1/generate", json={
            "tokens": [], "max_new_tokens": 4}).status_code == 400
        assert client.post("/v1/generate", json={
            "tokens": [1, 2], "max_new_tokens": 0}).status_code == 400

        s = client.get("/v1/stats").json()
        assert s["requests"] == 3 and s["tokens_generated"] == 6 + 5 + 5
        assert s["mean_batch"] >= 1.0
        assert s["latency_ms"]["p50"] > 0
        assert s["latency_ms"]["p95"] >= s["latency_ms"]["p50"]


@pytest.mark.gpu
def test_worker_gpu_batched_matches_reference():
    """Serving worker on MI355X: batched KV-cached decode through the
    HIP op path equals the direct Generator rollout."""
    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=128", "model.d_ff=256",
        "model.vocab_size=512", "model.max_seq_len=128",
    ])
    torch.manual_seed(11)
    m = build_model(cfg).to("cuda", torch.bfloat16).eval()
    w = InferenceWorker(m, max_batch=4, window_s=0.05)
    try:
        prompts = [[i, i + 1, i + 2, i + 3] for i in (1, 40, 300)]
        reqs = [w.submit(p, 16) for p in prompts]
        for r in reqs:
            assert r.event.wait(timeout=120) and r.error is None
        assert w.stats["batches"] == 1    # stacked into one decode call
        ref = Generator(m).generate(
            torch.tensor(prompts, dtype=torch.int64, device="cuda"), 16)
        for i, r in enumerate(reqs):
            assert r.result == ref[i, 4:].tolist()
    finally:
        w.close()


def test_text_completions_endpoint():
    from starlette.testclient import TestClient
    tokenizers = pytest.importorskip("tokenizers")
    from tokenizers import Tokenizer, models, pre_tokenizers, trainers

    tok = Tokenizer(models.BPE(unk_token="<unk>"))
    tok.pre_tokenizer = pre_tokenizers.Whitespace()
    corpus = ["the quick brown fox jumps over the lazy dog"] * 8
    tok.train_from_iterator(corpus, trainers.BpeTrainer(
        vocab_size=96, special_tokens=["<unk>"]))
    assert tok.get_vocab_size() <= 128  # fits the tiny model vocab

    app = create_app(_tiny_model(), window_s=0.0, tokenizer=tok)
    with TestClient(app) as client:
        r = client.post("/v1/completions", json={
            "prompt": "the quick brown fox", "max_new_tokens": 6})
        assert r.status_code == 200, r.text
        body = r.json()
        assert len(body["text"]) == 1 and isinstance(body["text"][0], str)
        assert len(body["tokens"][0]) == 6
        # batch of texts
        r2 = client.post("/v1/completions", json={
            "prompt": ["lazy dog", "quick fox"], "max_new_tokens": 4})
        assert r2.status_code == 200 and len(r2.json()["text"]) == 2
        assert client.post("/v1/completions", json={
            "prompt": "", "max_new_tokens": 4}).status_code == 400


def test_worker_eos_trim():
    m = _tiny_model()
    ref = Generator(m).generate(
        torch.tensor([[1, 2, 3, 4]], dtype=torch.int64), 8)[0, 4:].tolist()
    eos = ref[3]
    w = InferenceWorker(m, max_batch=4, window_s=0.0)
    try:
        r = w.submit([1, 2, 3, 4], 8, eos_token_id=eos)
        assert r.event.wait(30) and r.error is None
        assert r.result == ref[:3]
    finally:
        w.close()


def test_cli_weights_and_lora_flags(tmp_path):
    """serve CLI wiring: --weights + --lora produce a served model whose
    outputs equal the merged fine-tune (probed via create_app worker)."""
    import safetensors.torch as st

    from scaffold.convert import save_model
    from scaffold.lora import apply_lora, lora_state_dict, merge_lora
    from scaffold.models import build_model
    from scaffold.config import load_config

    over = [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
        "model.vocab_size=128", "model.max_seq_len=64",
    ]
    torch.manual_seed(81)
    base = build_model(load_config(None, over)).eval()
    wp = str(tmp_path / "w.safetensors")
    save_model(base, wp)
    apply_lora(base, r=4, alpha=8)
    with torch.no_grad():
        for p in base.parameters():
            if p.requires_grad:
                p.add_(torch.randn_like(p) * 0.03)
    ap = str(tmp_path / "a.safetensors")
    st.save_file(lora_state_dict(base), ap)
    merge_lora(base)

    # rebuild through the CLI path (main() minus uvicorn.run)
    import scaffold.serve as serve_mod

    captured = {}

    def fake_run(app, **kw):
        captured["app"] = app

    import uvicorn as _uv
    real_run = _uv.run
    _uv.run = fake_run
    try:
        serve_mod.main(over + ["--weights", wp, "--lora", ap,
                               "--lora-r", "4", "--lora-alpha", "8"])
    finally:
        _uv.run = real_run
    worker = captured["app"].state.worker
    try:
        ref = Generator(base).generate(
            torch.tensor([[1, 2, 3, 4]], dtype=torch.int64), 8)[0, 4:]
        r = worker.submit([1, 2, 3, 4], 8)
        assert r.event.wait(60) and r.error is None
        assert r.result == ref.tolist()
    finally:
        worker.close()


def test_worker_context_length_validation():
    m = _tiny_model()           # max_seq_len=64
    w = InferenceWorker(m, window_s=0.0)
    try:
        with pytest.raises(ValueError, match="exceeds"):
            w.submit([1] * 60, 10)
    finally:
        w.close()


def test_prometheus_metrics_endpoint():
    from starlette.testclient import TestClient

    app = create_app(_tiny_model(), window_s=0.0)
    with TestClient(app) as client:
        client.post("/v1/generate", json={"tokens": [1, 2, 3],
                                          "max_new_tokens": 4})
        r = client.get("/metrics")
        assert r.status_code == 200
        body = r.text
        assert "scaffold_requests_total 1" in body
        assert "scaffold_tokens_generated_total 4" in body
        assert 'scaffold_request_latency_ms{quantile="0.5"}' in body


def test_queue_bound_rejects_with_overload():
    from scaffold.serve import OverloadedError

    m = _tiny_model()
    w = InferenceWorker(m, window_s=5.0, max_queue=2)  # long window: queue holds
    try:
        w.submit([1, 2], 2)
        w.submit([3, 4], 2)
        with pytest.raises(OverloadedError):
            w.submit([5, 6], 2)
    finally:
        w.close()


def test_worker_speculative_path():
    """A worker with a draft model serves greedy B=1 requests through
    generate_speculative — results identical to plain decode."""
    torch.manual_seed(83)
    m = _tiny_model()
    cfg_d = load_config(None, [
        "model.name=llama", "model.n_layers=1", "model.n_heads=2",
        "model.n_kv_heads=1", "model.d_model=32", "model.d_ff=64",
        "model.vocab_size=128", "model.max_seq_len=64",
    ])
    torch.manual_seed(84)
    draft = build_model(cfg_d).eval()
    ref = Generator(m).generate(
        torch.tensor([[1, 2, 3, 4]], dtype=torch.int64), 10)[0, 4:].tolist()

    w = InferenceWorker(m, window_s=0.0, draft_model=draft, draft_k=3)
    try:
        r = w.submit([1, 2, 3, 4], 10)
        assert r.event.wait(60) and r.error is None
        assert r.result == ref
    finally:
        w.close()


@pytest.mark.timeout(120)
def test_real_uvicorn_server_over_tcp(tmp_path):
    """The stack outside TestClient: a real uvicorn server on a TCP
    port, queried with httpx."""
    import socket
    import threading
    import time as _time

    import httpx
    import uvicorn

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]

    app = create_app(_tiny_model(), window_s=0.0)
    config = uvicorn.Config(app, host="127.0.0.1", port=port,
                            log_level="error")
    server = uvicorn.Server(config)
    th = threading.Thread(target=server.run, daemon=True)
    th.start()
    try:
        deadline = _time.time() + 30
        url = f"http://127.0.0.1:{port}"
        while _time.time() < deadline:
            try:
                if httpx.get(url + "/healthz", timeout=2).status_code == 200:
                    break
            except Exception:
                _time.sleep(0.1)
        else:
            raise AssertionError("server never became healthy")
        r = httpx.post(url + "/v1/generate",
                       json={"tokens": [1, 2, 3], "max_new_tokens": 5},
                       timeout=30)
        assert r.status_code == 200, r.text
        assert len(r.json()["tokens"][0]) == 5
        assert httpx.get(url + "/metrics", timeout=10).status_code == 200
    finally:
        server.should_exit = True
        th.join(timeout=10)
