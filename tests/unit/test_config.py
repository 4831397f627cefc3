import json

import pytest

from scaffold.config import (TrainConfig, apply_overrides, config_from_dict,
                             config_hash, config_to_dict, load_config)


def test_defaults():
    cfg = TrainConfig()
    assert cfg.model.name == "mlp"
    assert cfg.dist.bucket_cap_mb == 25
    assert cfg.dtype == "bf16"


def test_roundtrip_dict():
    cfg = TrainConfig()
    cfg.model.name = "gpt2"
    cfg.optim.lr = 3e-4
    d = config_to_dict(cfg)
    cfg2 = config_from_dict(d)
    assert cfg2.model.name == "gpt2"
    assert cfg2.optim.lr == 3e-4
    assert config_hash(cfg) == config_hash(cfg2)


def test_overrides():
    cfg = TrainConfig()
    apply_overrides(cfg, ["optim.lr=0.01", "model.n_layers=4", "steps=7",
                          "dist.overlap=false", "dtype=fp32"])
    assert cfg.optim.lr == 0.01
    assert cfg.model.n_layers == 4
    assert cfg.steps == 7
    assert cfg.dist.overlap is False
    assert cfg.dtype == "fp32"


def test_unknown_key_rejected():
    cfg = TrainConfig()
    with pytest.raises(KeyError):
        apply_overrides(cfg, ["optim.nope=1"])
    with pytest.raises(KeyError):
        config_from_dict({"optim": {"nope": 1}})


def test_load_json_file(tmp_path):
    p = tmp_path / "c.json"
    p.write_text(json.dumps({"model": {"name": "resnet50"}, "steps": 3}))
    cfg = load_config(str(p), ["data.batch_size=8"])
    assert cfg.model.name == "resnet50"
    assert cfg.steps == 3
    assert cfg.data.batch_size == 8


def test_llama_preset():
    cfg = load_config(None, ["model.name=llama3_8b"])
    assert cfg.model.d_model == 4096
    assert cfg.model.n_kv_heads == 8
    assert cfg.model.d_ff == 14336
    assert cfg.model.vocab_size == 128256


def test_hash_changes_with_config():
    a = TrainConfig()
    b = TrainConfig()
    b.optim.lr = 123.0
    assert config_hash(a) != config_hash(b)


def test_bool_and_float_coercion():
    cfg = load_config(None, [
        "dist.overlap=false", "ckpt.async_d2h=0", "hip_graph=true",
        "optim.lr=2.5e-4", "dist.bucket_cap_mb=12.5",
    ])
    assert cfg.dist.overlap is False
    assert cfg.ckpt.async_d2h is False
    assert cfg.hip_graph is True
    assert abs(cfg.optim.lr - 2.5e-4) < 1e-12
    assert cfg.dist.bucket_cap_mb == 12.5


def test_latest_checkpoint_picks_newest_complete(tmp_path):
    import json
    import os

    from scaffold.ckpt import layout

    for step, complete in [(2, True), (10, True), (30, False)]:
        d = tmp_path / f"step_{step:08d}"
        d.mkdir()
        if complete:
            (d / "manifest.json").write_text(json.dumps({"step": step}))
    # newest COMPLETE is step 10 (step 30 lacks a manifest = mid-crash)
    got = layout.latest_checkpoint(str(tmp_path))
    assert got.endswith("step_00000010"), got
