"""CLI entrypoint, TOML config, and metrics-JSONL coverage."""

import json
import os
import subprocess
import sys

import torch

from scaffold.config import load_config
from scaffold.metrics import MetricLogger
from scaffold.runtime import DistContext

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def test_train_cli_smoke(tmp_path):
    """python train.py --config <toml> overrides... runs end to end."""
    cfgf = tmp_path / "t.toml"
    cfgf.write_text(
        'steps = 3\ndtype = "fp32"\ndevice = "cpu"\n'
        '[model]\nname = "mlp"\nin_dim = 16\nhidden_dim = 32\nout_dim = 4\n'
        '[data]\nkind = "synthetic_vector"\nbatch_size = 8\nvector_dim = 16\n'
    )
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "train.py"), "--config", str(cfgf),
         "log.interval=1", "optim.lr=0.01"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "samples/s" in out.stdout


def test_toml_and_json_configs_agree(tmp_path):
    j = tmp_path / "c.json"
    t = tmp_path / "c.toml"
    j.write_text('{"model": {"name": "gpt2", "n_layers": 3}, "steps": 7}')
    t.write_text('steps = 7\n[model]\nname = "gpt2"\nn_layers = 3\n')
    cj = load_config(str(j), [])
    ct = load_config(str(t), [])
    assert cj.model.n_layers == ct.model.n_layers == 3
    assert cj.steps == ct.steps == 7


def test_metrics_jsonl(tmp_path):
    mf = tmp_path / "m.jsonl"
    cfg = load_config(None, [f"log.metrics_file={mf}", "log.interval=1"])
    ctx = DistContext()
    lg = MetricLogger(ctx, cfg)
    lg.log_step(1, 2.5, samples=64, step_time_s=0.1, extra={"comm_ms": 1.25})
    lg.log_step(2, 2.0, samples=64, step_time_s=0.1)
    lg.close()
    recs = [json.loads(l) for l in mf.read_text().splitlines()]
    assert len(recs) == 2
    assert recs[0]["loss"] == 2.5 and recs[0]["comm_ms"] == 1.25
    assert recs[1]["samples_per_sec_node"] > 0
