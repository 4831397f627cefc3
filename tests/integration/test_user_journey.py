"""The full user journey through real entry points: pretrain ->
export -> LoRA fine-tune from those weights -> merge -> quantize ->
serve -> query.  Every arrow is a public CLI or API documented in
docs/TRAINING.md / docs/SERVING.md."""

import os
import subprocess
import sys

import numpy as np
import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

MODEL_OVER = [
    "model.name=llama", "model.n_layers=2", "model.n_heads=4",
    "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
    "model.vocab_size=128", "model.max_seq_len=64",
]


@pytest.mark.timeout(600)
def test_pretrain_finetune_serve_flow(tmp_path):
    import safetensors.torch as st

    from scaffold.config import load_config
    from scaffold.convert import save_model
    from scaffold.generate import Generator
    from scaffold.lora import lora_state_dict, merge_lora
    from scaffold.models import build_model
    from scaffold.quant import save_quantized
    from scaffold.serve import create_app
    from scaffold.trainer import Trainer

    corpus = tmp_path / "corpus.bin"
    np.random.default_rng(0).integers(
        0, 128, size=8192, dtype=np.uint16).tofile(corpus)

    # 1. pretrain via the real CLI (token-file data, checkpoints)
    ck = tmp_path / "ck"
    r = subprocess.run([
        sys.executable, os.path.join(REPO, "train.py"), *MODEL_OVER,
        "data.kind=token_file", f"data.path={corpus}",
        "data.batch_size=4", "data.seq_len=32", "dtype=fp32",
        "device=cpu", "log.interval=1000", "steps=3",
        f"ckpt.dir={ck}", "ckpt.save_final=true",
    ], capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-1500:]

    # 2. export single-file weights from the checkpoint
    from scaffold.ckpt import layout
    last = layout.latest_checkpoint(str(ck))
    cfg = load_config(None, MODEL_OVER + [
        f"ckpt.resume={last}", "device=cpu", "dtype=fp32"])
    t = Trainer(cfg)
    weights = str(tmp_path / "pretrained.safetensors")
    save_model(t.raw_model, weights)

    # 3. LoRA fine-tune FROM the exported weights (in-process Trainer)
    cfg_ft = load_config(None, MODEL_OVER + [
        f"ckpt.init_weights={weights}", "model.lora_r=4",
        "data.kind=token_file", f"data.path={corpus}",
        "data.batch_size=4", "data.seq_len=32", "dtype=fp32",
        "device=cpu", "log.interval=1000", "steps=5", "optim.lr=0.02",
    ])
    tf = Trainer(cfg_ft)
    tf.fit()
    adapters = str(tmp_path / "adapters.safetensors")
    st.save_file(lora_state_dict(tf.raw_model), adapters)

    # 4. merge + quantize for deployment
    merge_lora(tf.raw_model)
    deploy = str(tmp_path / "deploy_int8.safetensors")
    save_quantized(tf.raw_model, deploy)

    # 5. serve the quantized artifact and query it over HTTP
    from starlette.testclient import TestClient

    served = build_model(load_config(None, MODEL_OVER)).eval()
    from scaffold.quant import load_quantized
    load_quantized(served, deploy)
    app = create_app(served, continuous=True, max_batch=2)
    with TestClient(app) as client:
        resp = client.post("/v1/generate", json={
            "tokens": [1, 2, 3, 4], "max_new_tokens": 8})
        assert resp.status_code == 200, resp.text
        toks = resp.json()["tokens"][0]
        assert len(toks) == 8 and all(0 <= x < 128 for x in toks)
        # the served model is the fine-tune (int8-close): greedy rollout
        # matches direct decode of the deployed weights exactly
        ref = Generator(served).generate(
            torch.tensor([[1, 2, 3, 4]], dtype=torch.int64), 8)
        assert toks == ref[0, 4:].tolist()
