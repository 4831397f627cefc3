"""ws=2 rehearsal of bench.py's DISTRIBUTED path (CPU/gloo): the exact
contract the driver's multi-GPU scaling run exercises — env-var
rendezvous, barriers around the timed region, MAX-over-ranks elapsed,
rank-0-only JSON with whole-node value."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


@pytest.mark.timeout(300)
def test_bench_ws2_gloo(tmp_path):
    env_base = dict(os.environ)
    env_base.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": "29611",
        "WORLD_SIZE": "2",
    })
    procs = []
    for rank in range(2):
        env = dict(env_base)
        env["RANK"] = str(rank)
        env["LOCAL_RANK"] = str(rank)
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(REPO, "bench.py"), "--model", "mlp",
             "--steps", "4", "--warmup", "1"],
            env=env, cwd=REPO, stdout=subprocess.PIPE,
            stderr=subprocess.PIPE, text=True))
    outs = [p.communicate(timeout=240) for p in procs]
    for p, (o, e) in zip(procs, outs):
        assert p.returncode == 0, e[-2000:]
    jsons = [l for o, _ in outs for l in o.splitlines() if l.startswith("{")]
    assert len(jsons) == 1, "exactly rank 0 emits the JSON line"
    d = json.loads(jsons[0])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["value"] > 0
    # every field the driver's contract names must be present and typed
    for key, typ in (("metric", str), ("value", (int, float)),
                     ("unit", str), ("n_gpus", int), ("steps", int),
                     ("warmup", int), ("ms_per_step", (int, float)),
                     ("higher_is_better", bool), ("scaling", str),
                     ("dtype", str), ("data", str), ("config", dict)):
        assert key in d and isinstance(d[key], typ), key
    assert "vs_baseline" in d          # may be null
    assert d["scaling"] in ("weak", "strong")
    assert d["data"] == "synthetic"
    assert {"model", "global_batch"} <= set(d["config"])
