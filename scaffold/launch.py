"""Supervising elastic launcher: one process per GPU,
restart-from-last-checkpoint on rank failure.

The recovery model is deliberately minimal and honest: checkpoints are
written atomically (shards + fsync first, rank-0 manifest last —
scaffold/ckpt/layout.py), so the newest step directory WITH a manifest
is always a consistent world state.  When any rank dies (crash, OOM,
lost RCCL peer aborting on `dist.timeout_s`), this launcher kills the
remaining ranks BY EXACT PID, finds that directory with
`layout.latest_checkpoint`, and relaunches the whole world with
`ckpt.resume=<it>` — up to `--max-restarts` times.  Resume is bit-exact
(RNG/optimizer/data-cursor state live in the shards; proven by
tests/distrib/test_fault_recovery.py and test_launch_elastic.py).

Usage (8 × MI355X):
    python -m scaffold.launch --nproc 8 -- train.py --config cfg.json \
        ckpt.dir=/ckpts ckpt.interval=500
Each rank gets RANK/LOCAL_RANK/WORLD_SIZE/MASTER_ADDR=127.0.0.1 and a
fresh MASTER_PORT per attempt (a stale rendezvous port from the killed
world can not collide).
"""

from __future__ import annotations

import argparse
import os
import socket
import subprocess
import sys
import time
from typing import List, Optional

from .ckpt import layout


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _spawn_world(cmd: List[str], nproc: int, port: int,
                 extra_env: Optional[dict] = None) -> List[subprocess.Popen]:
    procs = []
    for rank in range(nproc):
        env = dict(os.environ)
        env.update({
            "RANK": str(rank), "LOCAL_RANK": str(rank),
            "WORLD_SIZE": str(nproc),
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        })
        if extra_env:
            env.update(extra_env)
        procs.append(subprocess.Popen(cmd, env=env))
    return procs


def _kill_world(procs: List[subprocess.Popen], grace_s: float = 5.0) -> None:
    """Exact-PID teardown: SIGTERM, then SIGKILL stragglers."""
    for p in procs:
        if p.poll() is None:
            p.terminate()
    deadline = time.time() + grace_s
    for p in procs:
        if p.poll() is None:
            try:
                p.wait(timeout=max(0.1, deadline - time.time()))
            except subprocess.TimeoutExpired:
                p.kill()
                p.wait()


def _watch(procs: List[subprocess.Popen], poll_s: float) -> int:
    """Block until the world ends.  0 = every rank exited cleanly;
    otherwise the first non-zero exit code seen (rest get killed)."""
    while True:
        codes = [p.poll() for p in procs]
        bad = [c for c in codes if c not in (None, 0)]
        if bad:
            _kill_world(procs)
            return bad[0]
        if all(c == 0 for c in codes):
            return 0
        time.sleep(poll_s)


def run_elastic(cmd: List[str], nproc: int, ckpt_dir: str,
                max_restarts: int = 3, poll_s: float = 0.2,
                stats: Optional[dict] = None) -> int:
    """Run `cmd` as an nproc-rank world; on failure restart from the
    last complete checkpoint in `ckpt_dir`.  Returns the final world
    exit code (0 on success).  `cmd` is the per-rank argv; on restart
    `ckpt.resume=<dir>` is appended (later key=value wins in the config
    loader, so a stale resume flag in cmd is overridden).  If `stats`
    is given, stats["restarts"] records how many worlds failed."""
    attempt = 0
    while True:
        attempt_cmd = list(cmd)
        resume = layout.latest_checkpoint(ckpt_dir) if attempt > 0 else ""
        if resume:
            attempt_cmd.append(f"ckpt.resume={resume}")
        # disarm the test-only fault hook on restarts (trainer.fit)
        extra = {"SCAFFOLD_FAULT_DONE": "1"} if attempt > 0 else None
        procs = _spawn_world(attempt_cmd, nproc, _free_port(), extra)
        code = _watch(procs, poll_s)
        if code == 0:
            if stats is not None:
                stats["restarts"] = attempt
            return 0
        attempt += 1
        print(f"[scaffold.launch] world failed (exit {code}), "
              f"attempt {attempt}/{max_restarts}, last good ckpt: "
              f"{layout.latest_checkpoint(ckpt_dir) or 'none'}",
              file=sys.stderr, flush=True)
        if attempt > max_restarts:
            if stats is not None:
                stats["restarts"] = attempt
            return code


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(
        description="elastic launcher: restart world from last checkpoint")
    ap.add_argument("--nproc", type=int, required=True)
    ap.add_argument("--max-restarts", type=int, default=3)
    ap.add_argument("--ckpt-dir", default="",
                    help="checkpoint base dir (default: parse ckpt.dir= "
                         "from the command's overrides)")
    ap.add_argument("cmd", nargs=argparse.REMAINDER,
                    help="-- <program> [args...] (rank env is injected)")
    args = ap.parse_args(argv)
    cmd = args.cmd[1:] if args.cmd and args.cmd[0] == "--" else args.cmd
    if not cmd:
        ap.error("missing command after --")
    ckpt_dir = args.ckpt_dir
    if not ckpt_dir:
        for a in cmd:
            if a.startswith("ckpt.dir="):
                ckpt_dir = a.split("=", 1)[1]
    if not ckpt_dir:
        ap.error("no --ckpt-dir and no ckpt.dir= override in the command")
    if not cmd[0].endswith(".py"):
        full = cmd
    else:
        full = [sys.executable] + cmd
    return run_elastic(full, args.nproc, ckpt_dir, args.max_restarts)


if __name__ == "__main__":
    sys.exit(main())
