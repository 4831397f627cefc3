"""The 'everything on' integration point: token-file data + activation
checkpointing + MoE FFN + grad clip + checkpoint/resume under ws=2
gloo DDP, end to end through real train.py subprocesses."""

import os
import subprocess
import sys

import numpy as np
import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


@pytest.mark.timeout(600)
def test_full_stack_ws2_train_resume(tmp_path):
    corpus = tmp_path / "corpus.bin"
    np.random.default_rng(0).integers(
        0, 128, size=8192, dtype=np.uint16).tofile(corpus)
    ck = tmp_path / "ck"

    def cmd(steps, resume=""):
        c = [
            sys.executable, os.path.join(REPO, "train.py"),
            "model.name=gpt2", "model.n_layers=2", "model.n_heads=2",
            "model.d_model=32", "model.vocab_size=128", "model.max_seq_len=16",
            "model.n_experts=4", "model.act_ckpt=true",
            "data.kind=token_file", f"data.path={corpus}",
            "data.batch_size=4", "data.seq_len=16",
            "optim.grad_clip=1.0", "dtype=fp32", "device=cpu",
            "log.interval=1000", f"ckpt.dir={ck}", "ckpt.interval=2",
            "ckpt.save_final=true", f"steps={steps}",
        ]
        if resume:
            c.append(f"ckpt.resume={resume}")
        return c

    def run_world(steps, resume="", port=0):
        import socket
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]
        procs = []
        for rank in range(2):
            env = dict(os.environ)
            env.update({"RANK": str(rank), "LOCAL_RANK": str(rank),
                        "WORLD_SIZE": "2", "MASTER_ADDR": "127.0.0.1",
                        "MASTER_PORT": str(port)})
            procs.append(subprocess.Popen(cmd(steps, resume), env=env))
        for p in procs:
            assert p.wait(timeout=300) == 0

    # 6 uninterrupted steps
    run_world(6)
    import safetensors.torch as st

    from scaffold.ckpt import layout
    last = layout.latest_checkpoint(str(ck))
    assert last.endswith("step_00000006")
    full = {}
    for f in sorted(os.listdir(last)):
        if f.startswith("shard_"):
            full.update(st.load_file(os.path.join(last, f)))
    ref = {k: v.clone() for k, v in full.items() if k.startswith("model.")}

    # 4 steps, then resume for 2 more in a fresh world
    ck2 = tmp_path / "ck2"
    ck = ck2
    run_world(4)
    mid = layout.latest_checkpoint(str(ck2))
    assert mid.endswith("step_00000004")
    run_world(6, resume=mid)
    last2 = layout.latest_checkpoint(str(ck2))
    assert last2.endswith("step_00000006")
    full2 = {}
    for f in sorted(os.listdir(last2)):
        if f.startswith("shard_"):
            full2.update(st.load_file(os.path.join(last2, f)))
    for k, v in ref.items():
        assert torch.equal(v, full2[k]), k
