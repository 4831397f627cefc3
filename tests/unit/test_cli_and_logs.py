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


def test_bench_sweep_buckets_cpu():
    """--sweep-buckets emits one JSON line per bucket size (§5.8 table
    in one launch) with the median/p10/p90 protocol fields."""
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--model", "mlp",
         "--steps", "3", "--warmup", "1", "--sweep-buckets", "0.125,0.25"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [json.loads(l) for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 2
    assert [l["config"]["bucket_cap_mb"] for l in lines] == [0.125, 0.25]
    for l in lines:
        assert l["sweep"] is True
        for k in ("ms_per_step_median", "ms_per_step_p10", "ms_per_step_p90"):
            assert k in l


def test_toml_and_json_configs_agree(tmp_path):
    j = tmp_path / "c.json"
    t = tmp_path / "c.toml"
    j.write_text('{"model": {"name": "gpt2", "n_layers": 3}, "steps": 7}')
    t.write_text('steps = 7\n[model]\nname = "gpt2"\nn_layers = 3\n')
    cj = load_config(str(j), [])
    ct = load_config(str(t), [])
    assert cj.model.n_layers == ct.model.n_layers == 3
    assert cj.steps == ct.steps == 7


def test_step_breakdown_fields_in_metrics(tmp_path):
    """§5.5: a fit() writes the data/fwd/bwd/opt breakdown into
    metrics.jsonl (device-event spans on GPU, host clock here)."""
    from scaffold.trainer import Trainer

    mf = tmp_path / "m.jsonl"
    cfg = load_config(None, [
        "model.name=mlp", "model.in_dim=16", "model.hidden_dim=32",
        "model.out_dim=4", "data.kind=synthetic_vector", "data.batch_size=8",
        "data.vector_dim=16", "dtype=fp32", "device=cpu", "steps=4",
        f"log.metrics_file={mf}", "log.interval=2",
    ])
    Trainer(cfg).fit()
    recs = [json.loads(l) for l in mf.read_text().splitlines()]
    assert recs, "no metrics written"
    for key in ("data_ms", "fwd_ms", "bwd_ms", "opt_ms"):
        assert key in recs[-1], (key, recs[-1])
        assert recs[-1][key] >= 0.0
    # phases roughly partition the step
    total = sum(recs[-1][k] for k in ("data_ms", "fwd_ms", "bwd_ms", "opt_ms"))
    assert total <= recs[-1]["ms_per_step"] * 3  # sanity, not exactness


def test_step_breakdown_bounded_memory():
    """drain() caps retained steps: a huge log interval must not
    accumulate events without bound."""
    from scaffold.metrics import StepBreakdown

    bd = StepBreakdown(torch.device("cpu"), keep=4)
    for _ in range(100):
        bd.start_step()
        bd.mark("fwd")
        bd.end_step()
    assert len(bd._steps) == 4
    out = bd.drain()
    assert "fwd_ms" in out and len(bd._steps) == 0


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


import pytest


@pytest.mark.parametrize("argv", [
    ["train.py", "--help"],
    ["bench.py", "--help"],
    ["-m", "scaffold.serve", "--help"],
    ["-m", "scaffold.launch", "--help"],
])
def test_cli_help_exits_zero(argv):
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    r = subprocess.run([sys.executable, *argv], cwd=repo,
                       capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr[-500:]
    assert "usage" in (r.stdout + r.stderr).lower()
