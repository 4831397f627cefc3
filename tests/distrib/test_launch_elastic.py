"""Elastic launcher end to end (scaffold/launch.py): real train.py
subprocess ranks on CPU/gloo, a rank dies uncleanly mid-run, the
launcher restarts the world from the last complete manifest, and the
final params are bit-exact with an uninterrupted run."""

import os
import subprocess
import sys

import numpy as np
import pytest
import torch

from scaffold.launch import run_elastic

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _cmd(ckpt_dir):
    return [
        sys.executable, os.path.join(REPO, "train.py"),
        "model.name=mlp", "model.in_dim=32", "model.hidden_dim=64",
        "model.out_dim=4", "data.kind=synthetic_vector",
        "data.batch_size=8", "data.vector_dim=32",
        "dtype=fp32", "device=cpu", "log.interval=1000",
        f"ckpt.dir={ckpt_dir}", "ckpt.interval=2", "ckpt.save_final=true",
        "steps=6",
    ]


def _final_params(ckpt_dir):
    import safetensors.torch as st

    from scaffold.ckpt import layout

    last = layout.latest_checkpoint(ckpt_dir)
    assert last, f"no checkpoint in {ckpt_dir}"
    state = {}
    for f in sorted(os.listdir(last)):
        if f.startswith("shard_") and f.endswith(".safetensors"):
            state.update(st.load_file(os.path.join(last, f)))
    keys = sorted(k for k in state if k.startswith("model."))
    assert keys
    return torch.cat([state[k].flatten().float() for k in keys]).numpy(), last


@pytest.mark.timeout(600)
def test_launcher_restarts_and_matches_uninterrupted(tmp_path):
    # run A: uninterrupted
    ck_a = str(tmp_path / "a")
    rc = run_elastic(_cmd(ck_a), nproc=2, ckpt_dir=ck_a, max_restarts=0)
    assert rc == 0
    pa, last_a = _final_params(ck_a)

    # run B: rank 1 dies at step 3 (after the step-2 checkpoint); the
    # launcher must notice, kill rank 0, and relaunch with resume
    ck_b = str(tmp_path / "b")
    os.environ["SCAFFOLD_FAULT_STEP"] = "3"
    os.environ["SCAFFOLD_FAULT_RANK"] = "1"
    stats = {}
    try:
        rc = run_elastic(_cmd(ck_b), nproc=2, ckpt_dir=ck_b, max_restarts=2,
                         stats=stats)
    finally:
        del os.environ["SCAFFOLD_FAULT_STEP"]
        del os.environ["SCAFFOLD_FAULT_RANK"]
    assert rc == 0
    assert stats["restarts"] >= 1, "fault was never injected"
    pb, last_b = _final_params(ck_b)

    assert os.path.basename(last_a) == os.path.basename(last_b)
    assert np.array_equal(pa, pb)


@pytest.mark.timeout(300)
def test_launcher_gives_up_after_max_restarts(tmp_path):
    ck = str(tmp_path / "c")
    cmd = [sys.executable, "-c", "import sys; sys.exit(3)"]
    rc = run_elastic(cmd, nproc=2, ckpt_dir=ck, max_restarts=1)
    assert rc == 3


def test_cli_requires_ckpt_dir():
    from scaffold.launch import main

    with pytest.raises(SystemExit):
        main(["--nproc", "2", "--", "train.py", "steps=1"])
