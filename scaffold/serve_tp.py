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

    def submit(self, tokens, max_new_tokens, temperature=0.0, **kw):
        if temperature > 0:
            raise ValueError("TP serving is greedy-only (per-rank "
                             "sampling RNG would diverge)")
        return super().submit(tokens, max_new_tokens, temperature, **kw)

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
