"""Tensor-parallel serving: rank 0 runs the HTTP front end and the
continuous batcher; follower ranks replay the same engine steps on
their weight shards so every rank contributes 1/world of each block's
GEMV work (shard layout: dist.tp.shard_llama_tp).

Protocol (greedy-only — followers recompute the same argmax from the
identical post-all_reduce logits, so NO token traffic is needed):

  rank 0's engine thread broadcasts a small command tensor before every
  engine action; followers loop on it:
    [PREFILL, slot, prompt_len, max_new]  + a prompt broadcast
    [DECODE]                               (one varpos step, all slots)
    [STOP]
  Both sides then run IDENTICAL code paths — the TP all_reduces inside
  the sharded forward are the only other collectives, and they match
  because the engine action sequence matches.  Slot bookkeeping
  (admission order, counts, retirement) is deterministic given the
  command stream, so the mirrors never diverge.

Sampled requests are rejected at submit (per-rank RNG would diverge);
streaming/prefix-cache work as on a single rank (rank 0 owns results;
followers keep their own consistent prefix caches).

ws=2 gloo exactness: tests/distrib/test_serve_tp.py — sharded TP
serving == unsharded rollout token-for-token.  Multi-GPU RCCL timing
is a round-3 item (NOTES roadmap).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from .serve import ContinuousBatcher

_PREFILL, _DECODE, _STOP = 0, 1, 2


class TpContinuousBatcher(ContinuousBatcher):
    """Rank-0 side: a ContinuousBatcher whose engine actions are
    announced to the follower ranks first."""

    def __init__(self, model, group=None, **kw):
        kw.setdefault("use_graph", False)
        self._tp_pgroup = group
        super().__init__(model, **kw)

    def submit(self, tokens, max_new_tokens, temperature=0.0,
               top_k=0, top_p=0.0, **kw):
        if temperature > 0:
            raise ValueError("TP serving is greedy-only (per-rank "
                             "sampling RNG would diverge)")
        return super().submit(tokens, max_new_tokens, temperature,
                              top_k, top_p, **kw)

    def _announce(self, payload):
        t = torch.tensor(payload, dtype=torch.int64)
        dist.broadcast(t, src=0, group=self._tp_pgroup)

    def _admit(self, slot, req):
        self._announce([_PREFILL, slot, int(req.tokens.numel()),
                        req.max_new_tokens])
        dist.broadcast(req.tokens, src=0, group=self._tp_pgroup)
        super()._admit(slot, req)

    def _step(self):
        self._announce([_DECODE, 0, 0, 0])
        super()._step()

    def _on_stop(self):
        self._announce([_STOP, 0, 0, 0])


class _FollowerRequest:
    """Minimal stand-in carrying what the engine bookkeeping reads."""

    def __init__(self, tokens, max_new_tokens):
        self.tokens = tokens
        self.max_new_tokens = max_new_tokens
        self.temperature = 0.0
        self.top_k = 0
        self.top_p = 0.0
        self.eos_token_id = None
        self.t_submit = 0.0
        self.stream = None
        self.stream_every = 4
        self.result = None
        self.error = None

    def trim_eos(self, toks):
        return toks

    class _Ev:
        def set(self):
            pass

    event = _Ev()


class _Follower(ContinuousBatcher):
    """Follower side: same engine, no queue thread — steps are driven
    by the command stream."""

    def __init__(self, model, group=None, **kw):
        kw.setdefault("use_graph", False)
        self._tp_pgroup = group
        super().__init__(model, **kw)
        # the base class started a queue thread; stop it (commands, not
        # the queue, drive this instance) — safe: no requests queued yet
        with self._cv:
            self._stop = True
            self._cv.notify()
        self._thread.join(timeout=10)
        self._stop = False   # re-arm engine methods

    def _loop(self):  # pragma: no cover - exits immediately via _stop
        return

    def serve_commands(self):
        dev = self.device
        while True:
            cmd = torch.zeros(4, dtype=torch.int64)
            dist.broadcast(cmd, src=0, group=self._tp_pgroup)
            kind = int(cmd[0])
            if kind == _STOP:
                return
            if kind == _PREFILL:
                slot, plen, max_new = int(cmd[1]), int(cmd[2]), int(cmd[3])
                toks = torch.zeros(plen, dtype=torch.int64)
                dist.broadcast(toks, src=0, group=self._tp_pgroup)
                self._ensure_cache()
                ContinuousBatcher._admit(
                    self, slot, _FollowerRequest(toks.to(dev), max_new))
                st = self._slots[slot]
                if st["count"] >= st["req"].max_new_tokens:
                    self._finish(slot)
            elif kind == _DECODE:
                ContinuousBatcher._step(self)


def run_follower(model, group=None, **kw):
    """Blocking follower loop for ranks > 0 (returns on STOP)."""
    _Follower(model, group=group, **kw).serve_commands()


def main(argv=None):
    """Launch TP serving under a multi-process launcher:

        torchrun --nproc-per-node 2 -m scaffold.serve_tp \\
            model.name=llama3_8b --port 8000

    Rank 0 serves HTTP over its shard; the other ranks follow.  Uses
    the same config overrides / --weights flags as scaffold.serve."""
    import argparse
    import os

    from .config import load_config
    from .dist.tp import shard_llama_tp
    from .models import build_model

    ap = argparse.ArgumentParser(description="tensor-parallel serving")
    ap.add_argument("overrides", nargs="*")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--max-slots", type=int, default=8)
    ap.add_argument("--weights", default="")
    args = ap.parse_args(argv)

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29531")
    dist.init_process_group(backend, rank=rank, world_size=world)
    try:
        dev = "cuda" if torch.cuda.is_available() else "cpu"
        if dev == "cuda":
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
        dt = torch.bfloat16 if dev == "cuda" else torch.float32
        cfg = load_config(None, args.overrides)
        torch.manual_seed(cfg.seed)     # identical init on every rank
        model = build_model(cfg)
        if args.weights:
            from .trainer import Trainer

            Trainer._load_init_weights(model, args.weights)
        model = model.to(dev, dt).eval()
        shard_llama_tp(model, rank, world)
        if world == 1 or rank == 0:
            import uvicorn

            from .serve import create_app

            worker = (TpContinuousBatcher(model, max_slots=args.max_slots)
                      if world > 1 else None)
            app = create_app(model, continuous=True,
                             max_batch=args.max_slots, worker=worker)
            uvicorn.run(app, host=args.host, port=args.port,
                        log_level="warning")
        else:
            run_follower(model, max_slots=args.max_slots)
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
