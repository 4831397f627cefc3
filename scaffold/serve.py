"""HTTP serving for the scaffold's LM families (guide: docs/SERVING.md).

Two engines behind one submit() surface:

- `InferenceWorker` — same-shape dynamic batching: requests with equal
  (prompt length, sampling params, eos) arriving within a short window
  are stacked into ONE `Generator.generate` call.  Grouping by prompt
  length keeps the batch un-padded (the cache layout has no padding
  mask, so mixed lengths would silently change positions).
- `ContinuousBatcher` — vLLM-style slot pool over the variable-position
  decode step: per-slot admission/retirement mid-decode, device-resident
  slot state (zero host syncs per step), optional hipGraph-replayed
  greedy step, exact-prefix KV reuse on admission, SSE streaming.

MI355X notes: decode is weight-read-bound (NOTES.md r2 serving ladder,
profiles/decode_batch_sweep_r2.txt: step time flat in batch size), so
slot occupancy converts directly into tokens/s — measured 2.03x serial
at 8 slots (profiles/serve_continuous_r2.txt).

Endpoints (see docs/SERVING.md for bodies): POST /v1/generate (ids,
optional SSE stream), POST /v1/completions (text, with --tokenizer),
GET /healthz, GET /v1/stats, GET /metrics (Prometheus).

Run: `python -m scaffold.serve model.name=llama3_8b [--continuous]
[--graph] [--weights w.safetensors] [--lora a.safetensors --lora-r 8]`
"""

from __future__ import annotations

import asyncio
import threading
import time
from collections import deque
from dataclasses import dataclass, field
from typing import List, Optional

import torch

from .generate import Generator

# batch assembly: requests arriving within this window join the batch
DEFAULT_WINDOW_S = 0.002

# bounded per-worker latency history for the stats endpoint
LATENCY_HISTORY = 512


class OverloadedError(RuntimeError):
    """Queue full — the HTTP layer maps this to 503."""


def _percentile(sorted_vals, q):
    if not sorted_vals:
        return 0.0
    i = min(len(sorted_vals) - 1, int(q * (len(sorted_vals) - 1) + 0.5))
    return sorted_vals[i]

try:
    from pydantic import BaseModel as _BaseModel

    # module scope: `from __future__ import annotations` stringifies the
    # endpoint annotation, and FastAPI can only resolve it against module
    # globals (a function-local class silently degrades to a query param)
    class GenRequest(_BaseModel):
        tokens: list       # [int] or [[int]]
        max_new_tokens: int = 32
        temperature: float = 0.0
        top_k: int = 0
        top_p: float = 0.0
        eos_token_id: "int | None" = None
        stream: bool = False

    class TextRequest(_BaseModel):
        prompt: object             # str or [str]
        max_new_tokens: int = 32
        temperature: float = 0.0
        top_k: int = 0
        top_p: float = 0.0
        eos_token_id: "int | None" = None
except ImportError:        # pydantic optional: InferenceWorker still works
    GenRequest = None
    TextRequest = None


@dataclass
class _Request:
    tokens: torch.Tensor            # (S,) int64 CPU
    max_new_tokens: int
    temperature: float
    top_k: int = 0
    top_p: float = 0.0
    eos_token_id: "int | None" = None
    t_submit: float = 0.0
    # streaming (continuous batcher only): sampled ids are pushed to
    # this queue in chunks of `stream_every` tokens, then a None
    # sentinel; the final `result` is still set at finish
    stream: "object | None" = None
    stream_every: int = 4
    event: threading.Event = field(default_factory=threading.Event)
    result: Optional[List[int]] = None
    error: Optional[Exception] = None

    @property
    def key(self):
        return (int(self.tokens.numel()), round(float(self.temperature), 6),
                self.top_k, round(float(self.top_p), 6), self.eos_token_id)

    def trim_eos(self, toks):
        """Cut at the first EOS (exclusive)."""
        if self.eos_token_id is None or self.eos_token_id not in toks:
            return toks
        return toks[:toks.index(self.eos_token_id)]


class InferenceWorker:
    """Owns the model; one thread serializes all GPU work.

    submit() is thread-safe and returns the _Request whose `event`
    fires when `result` (list of generated token ids, prompt excluded)
    or `error` is set.
    """

    def __init__(self, model: torch.nn.Module, max_batch: int = 8,
                 window_s: float = DEFAULT_WINDOW_S, use_graph: bool = False,
                 max_new_tokens_cap: int = 512, max_queue: int = 256,
                 draft_model: "torch.nn.Module | None" = None,
                 draft_k: int = 4):
        self.max_queue = max_queue
        self.gen = Generator(model)
        # speculative decoding (greedy B=1 requests only — per-row
        # acceptance lengths diverge in a batch): a small draft model
        # proposes draft_k tokens per target verify
        self.draft = Generator(draft_model.eval()) if draft_model is not None \
            else None
        self.draft_k = draft_k
        self.max_len = (model.rope_cos.shape[0] if self.gen.is_llama
                        else model.wpe.num_embeddings)
        self.device = next(model.parameters()).device
        self.max_batch = max_batch
        self.window_s = window_s
        self.use_graph = use_graph
        self.max_new_tokens_cap = max_new_tokens_cap
        self._q: deque[_Request] = deque()
        self._cv = threading.Condition()
        self._stop = False
        self.stats = {"requests": 0, "batches": 0, "tokens_generated": 0,
                      "started_at": time.time()}
        self.latencies_ms: deque = deque(maxlen=LATENCY_HISTORY)
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="scaffold-serve-worker")
        self._thread.start()

    def submit(self, tokens: List[int], max_new_tokens: int,
               temperature: float = 0.0, top_k: int = 0,
               top_p: float = 0.0, eos_token_id=None) -> _Request:
        if not tokens:
            raise ValueError("empty prompt")
        n = min(int(max_new_tokens), self.max_new_tokens_cap)
        if n < 1:
            raise ValueError("max_new_tokens must be >= 1")
        if len(tokens) + n > self.max_len:
            raise ValueError(
                f"prompt+max_new_tokens {len(tokens) + n} exceeds "
                f"max_len {self.max_len}")
        req = _Request(torch.tensor(tokens, dtype=torch.int64), n,
                       float(temperature), int(top_k), float(top_p),
                       eos_token_id, t_submit=time.time())
        with self._cv:
            if self._stop:
                raise RuntimeError("worker is shut down")
            if len(self._q) >= self.max_queue:
                raise OverloadedError(
                    f"queue full ({self.max_queue} pending)")
            self._q.append(req)
            self._cv.notify()
        return req

    def close(self):
        with self._cv:
            self._stop = True
            self._cv.notify()
        self._thread.join(timeout=5)
        # no waiter may hang on a request the dead worker will never run
        with self._cv:
            while self._q:
                r = self._q.popleft()
                r.error = RuntimeError("worker shut down")
                r.event.set()

    # -- worker side --------------------------------------------------

    def _take_batch(self) -> List[_Request]:
        with self._cv:
            while not self._q and not self._stop:
                self._cv.wait()
            if self._stop and not self._q:
                return []
            head = self._q.popleft()
        if self.window_s > 0:
            time.sleep(self.window_s)   # let same-shape peers arrive
        batch = [head]
        with self._cv:
            keep = deque()
            while self._q and len(batch) < self.max_batch:
                r = self._q.popleft()
                if r.key == head.key:
                    batch.append(r)
                else:
                    keep.append(r)
            keep.extend(self._q)
            self._q = keep
        return batch

    @torch.no_grad()
    def _run_batch(self, batch: List[_Request]):
        prompt = torch.stack([r.tokens for r in batch]).to(self.device)
        steps = max(r.max_new_tokens for r in batch)
        temp = batch[0].temperature
        eos = batch[0].eos_token_id
        if (self.draft is not None and len(batch) == 1 and temp == 0.0):
            out = self.gen.generate_speculative(
                prompt, steps, self.draft, k=self.draft_k,
                eos_token_id=eos)
        elif self.use_graph and temp == 0.0 and eos is None:
            out = self.gen.generate_graphed(prompt, steps)
        else:
            out = self.gen.generate(prompt, steps, temperature=temp,
                                    top_k=batch[0].top_k,
                                    top_p=batch[0].top_p,
                                    eos_token_id=eos)
        s0 = prompt.shape[1]
        new = out[:, s0:].cpu()
        now = time.time()
        for i, r in enumerate(batch):
            r.result = r.trim_eos(new[i, :r.max_new_tokens].tolist())
            self.stats["tokens_generated"] += len(r.result)
            self.latencies_ms.append((now - r.t_submit) * 1e3)
        self.stats["batches"] += 1
        self.stats["requests"] += len(batch)

    def _loop(self):
        while True:
            batch = self._take_batch()
            if not batch:
                return
            try:
                self._run_batch(batch)
            except Exception as e:          # surface, don't kill the loop
                for r in batch:
                    r.error = e
            for r in batch:
                r.event.set()


class ContinuousBatcher:
    """vLLM-style continuous batching over the Generator's variable-
    position decode step: a fixed pool of KV-cache SLOTS, each at its
    own depth; finished requests retire and new ones are admitted
    mid-decode without draining the batch.

    Per-step work is one decode_step_varpos over all slots — on MI355X
    the step is weight-read-bound (profiles/decode_batch_sweep_r2.txt:
    step time flat in batch size), so keeping slots full converts
    directly into tokens/s.  Admission runs a (dynamic-shape) prefill
    for the new request and copies its kv rows into the slot.

    API-compatible with InferenceWorker.submit; per-slot greedy or
    sampled decoding (temperature/top_k/top_p per request)."""

    def __init__(self, model: torch.nn.Module, max_slots: int = 8,
                 max_len: int = 0, max_new_tokens_cap: int = 512,
                 seed: int = 0, use_graph: bool = False,
                 prefix_cache_entries: int = 8, max_queue: int = 256):
        self.max_queue = max_queue
        self.gen = Generator(model.eval())
        self.device = next(model.parameters()).device
        m = model
        self.is_llama = self.gen.is_llama
        attn0 = m.blocks[0].attn
        self.n_kv = attn0.n_kv if self.is_llama else attn0.n_heads
        self.head_dim = attn0.head_dim
        self.n_layers = len(m.blocks)
        if not max_len:
            max_len = (m.rope_cos.shape[0] if self.is_llama
                       else m.wpe.num_embeddings)
        self.max_len = max_len
        self.max_slots = max_slots
        self.max_new_tokens_cap = max_new_tokens_cap
        self.dtype = next(m.parameters()).dtype
        self._cache = None              # allocated on first request
        self._slots = [None] * max_slots  # per-slot state dicts
        # device-resident decode state: current token and generated-token
        # ring per slot.  Keeping these on device means a decode step
        # costs ZERO host syncs (the 1.76x-serial first cut synced 8x
        # per step pulling sampled ids back; see NOTES r2 serving) —
        # token values only come host-side once, at request finish.
        self._cur = torch.zeros(max_slots, 1, dtype=torch.int64,
                                device=self.device)
        # +1 scratch column: the static step scatters INACTIVE rows'
        # (ignored) tokens there so the kernel stays branch-free
        self._toks = torch.zeros(max_slots, max_new_tokens_cap + 1,
                                 dtype=torch.int64, device=self.device)
        self._pos = torch.zeros(max_slots, dtype=torch.int64,
                                device=self.device)
        self._count = torch.zeros(max_slots, dtype=torch.int64,
                                  device=self.device)
        self._active = torch.zeros(max_slots, dtype=torch.int64,
                                   device=self.device)
        self.use_graph = use_graph
        self._graph = None
        # exact-prefix KV cache: completed prefills keyed by their token
        # tuple; a new request reuses the longest cached prefix of its
        # prompt and prefills only the tail (system-prompt reuse).  LRU
        # by entry count; entries hold per-layer (k, v) host... kept on
        # DEVICE (serving has 288 GB of HBM; an 8-entry x 4k-token llama
        # cache is ~2 GB)
        self._prefix_entries = max(0, int(prefix_cache_entries))
        self._prefix_cache = {}          # tuple(tokens) -> [(k, v) x layer]
        self._prefix_lru = deque()
        self._q: deque[_Request] = deque()
        self._cv = threading.Condition()
        self._stop = False
        self._sample_gen = torch.Generator(device=self.device)
        self._sample_gen.manual_seed(seed)
        self.stats = {"requests": 0, "steps": 0, "tokens_generated": 0,
                      "started_at": time.time()}
        self.latencies_ms: deque = deque(maxlen=LATENCY_HISTORY)
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="scaffold-cont-batcher")
        self._thread.start()

    # InferenceWorker-compatible surface ------------------------------
    def submit(self, tokens, max_new_tokens: int, temperature: float = 0.0,
               top_k: int = 0, top_p: float = 0.0,
               eos_token_id=None, stream=None,
               stream_every: int = 4) -> _Request:
        if not tokens:
            raise ValueError("empty prompt")
        n = min(int(max_new_tokens), self.max_new_tokens_cap)
        if n < 1:
            raise ValueError("max_new_tokens must be >= 1")
        if len(tokens) + n > self.max_len:
            raise ValueError(
                f"prompt+max_new_tokens {len(tokens) + n} exceeds "
                f"max_len {self.max_len}")
        req = _Request(torch.tensor(tokens, dtype=torch.int64), n,
                       float(temperature), int(top_k), float(top_p),
                       eos_token_id, t_submit=time.time(),
                       stream=stream, stream_every=max(1, int(stream_every)))
        with self._cv:
            if self._stop:
                raise RuntimeError("batcher is shut down")
            if len(self._q) >= self.max_queue:
                raise OverloadedError(
                    f"queue full ({self.max_queue} pending)")
            self._q.append(req)
            self._cv.notify()
        return req

    def close(self):
        with self._cv:
            self._stop = True
            self._cv.notify()
        self._thread.join(timeout=10)

    # worker side ------------------------------------------------------
    def _ensure_cache(self):
        if self._cache is None:
            from .generate import KVCache
            self._cache = KVCache(self.n_layers, self.max_slots, self.n_kv,
                                  self.max_len, self.head_dim, self.device,
                                  self.dtype)

    def _prefix_lookup(self, toks: tuple):
        """Longest cached strict prefix of `toks` (or `toks` itself)."""
        best = None
        for key in self._prefix_cache:
            if len(key) <= len(toks) and toks[:len(key)] == key:
                if best is None or len(key) > len(best):
                    best = key
        return best

    def _prefix_store(self, toks: tuple, kv):
        if self._prefix_entries == 0 or toks in self._prefix_cache:
            return
        while len(self._prefix_lru) >= self._prefix_entries:
            old = self._prefix_lru.popleft()
            self._prefix_cache.pop(old, None)
        self._prefix_cache[toks] = kv
        self._prefix_lru.append(toks)

    @torch.no_grad()
    def _admit(self, slot: int, req: _Request):
        """Prefill the request and install it in `slot`; a cached
        prefix of the prompt skips its share of the prefill."""
        from .generate import KVCache

        prompt = req.tokens.view(1, -1).to(self.device)
        S0 = prompt.shape[1]
        toks = tuple(int(t) for t in req.tokens)
        hit = self._prefix_lookup(toks)
        tmp = KVCache(self.n_layers, 1, self.n_kv, S0, self.head_dim,
                      self.device, self.dtype)
        if hit is not None and len(hit) < S0:
            # seed the cache with the reused prefix, prefill the tail
            for i, (hk, hv) in enumerate(self._prefix_cache[hit]):
                tmp.k[i][0, :, :len(hit)] = hk
                tmp.v[i][0, :, :len(hit)] = hv
            tmp.len = len(hit)
            logits = self.gen._forward_tokens(prompt[:, len(hit):], tmp,
                                              len(hit))
            self.stats["prefix_hits"] = self.stats.get("prefix_hits", 0) + 1
            self.stats["prefix_tokens_reused"] = (
                self.stats.get("prefix_tokens_reused", 0) + len(hit))
        elif hit is not None:
            # full-prompt hit still needs last-token logits: reuse all
            # but the final token, run one-step prefill for it
            n = len(hit) - 1
            for i, (hk, hv) in enumerate(self._prefix_cache[hit]):
                tmp.k[i][0, :, :n] = hk[:, :n]
                tmp.v[i][0, :, :n] = hv[:, :n]
            tmp.len = n
            logits = self.gen._forward_tokens(prompt[:, n:], tmp, n)
            self.stats["prefix_hits"] = self.stats.get("prefix_hits", 0) + 1
            self.stats["prefix_tokens_reused"] = (
                self.stats.get("prefix_tokens_reused", 0) + n)
        else:
            logits = self.gen._forward_tokens(prompt, tmp, 0)
        if self._prefix_entries:
            self._prefix_store(
                toks, [(tmp.k[i][0, :, :S0].clone(),
                        tmp.v[i][0, :, :S0].clone())
                       for i in range(self.n_layers)])
        for i in range(self.n_layers):
            self._cache.k[i][slot, :, :S0] = tmp.k[i][0]
            self._cache.v[i][slot, :, :S0] = tmp.v[i][0]
        first = Generator._sample(logits, req.temperature, req.top_k,
                                  req.top_p, self._sample_gen)
        self._cur[slot].copy_(first[0])
        self._toks[slot, 0].copy_(first[0, 0])
        self._slots[slot] = {"req": req, "count": 1, "pos": S0}

    def _finish(self, slot: int):
        st = self._slots[slot]
        # the one host sync of a request's lifetime; EOS is trimmed at
        # harvest (the slot runs to max_new_tokens — early slot freeing
        # would need a per-step device read, defeating the sync-free
        # step)
        st["req"].result = st["req"].trim_eos(
            self._toks[slot, :st["count"]].cpu().tolist())
        self.latencies_ms.append((time.time() - st["req"].t_submit) * 1e3)
        self.stats["tokens_generated"] += len(st["req"].result)
        self.stats["requests"] += 1
        st["req"].event.set()
        self._slots[slot] = None

    @torch.no_grad()
    def _step_static(self, hi: Optional[int] = None):
        """The STATIC greedy step: reads/writes only the persistent
        device buffers (_cur/_pos/_count/_active/_toks/cache), no host
        reads — hipGraph-capturable (use_graph=True replays the
        full-width form).  `hi` restricts the step to the first `hi`
        slots (zero-copy views; eager path only)."""
        if hi is None:
            hi = self.max_slots
            cache = self._cache
        else:
            cache = self._SlicedCache(self._cache, hi)
        cur, pos = self._cur[:hi], self._pos[:hi]
        count, act1 = self._count[:hi], self._active[:hi]
        logits = self.gen.decode_step_varpos(cur, cache, pos)
        nxt = logits.argmax(dim=-1, keepdim=True)
        new = torch.where(act1.unsqueeze(1).bool(), nxt, cur)
        cur.copy_(new)
        # active rows land at their count; inactive rows at the scratch
        # column (cap)
        cap = self._toks.shape[1] - 1
        idx = torch.where(act1.bool(), count, torch.full_like(count, cap))
        self._toks[:hi].scatter_(1, idx.unsqueeze(1), new)
        count.add_(act1)
        pos.add_(act1)

    def _upload_state(self):
        """Refresh the device buffers from the host slot mirrors (tiny
        H2D copies, no device sync); the host stays the source of truth
        so the eager sampled path and the static path can interleave."""
        host_pos, host_cnt, host_act = [], [], []
        for st in self._slots:
            host_pos.append(st["pos"] if st is not None else self.max_len - 1)
            host_cnt.append(st["count"] if st is not None else 0)
            host_act.append(1 if st is not None else 0)
        self._pos.copy_(torch.tensor(host_pos, dtype=torch.int64))
        self._count.copy_(torch.tensor(host_cnt, dtype=torch.int64))
        self._active.copy_(torch.tensor(host_act, dtype=torch.int64))

    def _flush_stream(self, b, st, final=False):
        """Push newly generated ids to the request's stream queue.
        Costs ONE device read per flush — streaming requests trade a
        sync every `stream_every` tokens for liveness; non-streaming
        slots keep the sync-free step."""
        r = st["req"]
        if r.stream is None or st.get("stream_done"):
            return
        seen = st.get("seen", 0)
        if not final and (st["count"] - seen) < r.stream_every:
            return
        if st["count"] > seen:
            chunk = self._toks[b, seen:st["count"]].cpu().tolist()
            st["seen"] = st["count"]
            if r.eos_token_id is not None and r.eos_token_id in chunk:
                chunk = chunk[:chunk.index(r.eos_token_id)]
                st["stream_done"] = True
            for t in chunk:
                r.stream.put(t)
        # the sentinel MUST go out on final even if nothing new was
        # generated since the last flush — an SSE reader blocks on it
        if final or st.get("stream_done"):
            r.stream.put(None)
            st["stream_done"] = True

    def _advance_host(self):
        for b, st in enumerate(self._slots):
            if st is None:
                continue
            st["pos"] += 1
            st["count"] += 1
            if st["count"] >= st["req"].max_new_tokens:
                self._flush_stream(b, st, final=True)
                self._finish(b)
            else:
                self._flush_stream(b, st)

    class _SlicedCache:
        """Zero-copy view of the first `hi` slots of a KVCache (the
        varpos step only needs .k/.v row views)."""

        def __init__(self, cache, hi):
            self.k = [t[:hi] for t in cache.k]
            self.v = [t[:hi] for t in cache.v]

    @torch.no_grad()
    def _step(self):
        """One decode step over every occupied slot — no host syncs:
        positions/counts are host ints (host->device copies only),
        sampled ids stay in the device-side _cur/_toks buffers.

        Eager steps compute only the OCCUPIED PREFIX of the slot pool
        (admission always picks the lowest free index, so a half-empty
        pool costs half a step, not a full one); the graphed step keeps
        the full static width so ONE capture serves every occupancy."""
        active = [b for b, st in enumerate(self._slots) if st is not None]
        greedy = all(self._slots[b]["req"].temperature <= 0 for b in active)
        self._upload_state()
        if greedy and self.use_graph and self.device.type == "cuda":
            self._graphed_step()
            self.stats["steps"] += 1
            self._advance_host()
            return
        hi = max(active) + 1
        if greedy:
            self._step_static(hi)
            self.stats["steps"] += 1
            self._advance_host()
            return
        # sampled path (per-request temperature/top_k/top_p): eager
        logits = self.gen.decode_step_varpos(
            self._cur[:hi], self._SlicedCache(self._cache, hi),
            self._pos[:hi])
        self.stats["steps"] += 1
        nxt = self._cur[:hi].clone()
        for b in active:
            r = self._slots[b]["req"]
            nxt[b] = Generator._sample(logits[b:b + 1], r.temperature,
                                       r.top_k, r.top_p,
                                       self._sample_gen)[0]
        self._cur[:hi].copy_(nxt)
        for b in active:
            self._toks[b, self._slots[b]["count"]].copy_(nxt[b, 0])
        self._advance_host()

    def _on_stop(self):
        """Engine-thread shutdown hook (see _loop); default no-op."""

    def _graphed_step(self):
        """Capture once (after two warmup executions — generate_graphed's
        proven recipe), then replay; the graph reads the persistent
        buffers that _upload_state refreshes.  TP-sharded models fall
        back to the eager step: capturing the in-step all_reduces is
        unvalidated on RCCL."""
        if getattr(self.gen.m.blocks[0].attn, "_tp_world", 1) > 1:
            self._step_static()
            return
        if self._graph is None:
            if getattr(self, "_warmups", 0) < 2:
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    self._step_static()
                torch.cuda.current_stream().wait_stream(side)
                self._warmups = getattr(self, "_warmups", 0) + 1
                return
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._step_static()   # records WITHOUT executing
            self._graph = g
        self._graph.replay()

    def _loop(self):
        while True:
            with self._cv:
                while (not self._q
                       and not any(s is not None for s in self._slots)
                       and not self._stop):
                    self._cv.wait()
                if self._stop:
                    # subclass hook (TP serving broadcasts STOP to the
                    # follower ranks from HERE — the engine thread — so
                    # it is ordered after any in-flight step's
                    # collectives)
                    self._on_stop()
                    # fail everything still queued or mid-decode so no
                    # waiter hangs on a worker that will never run again
                    err = RuntimeError("batcher shut down")
                    for r in self._q:
                        r.error = err
                        if r.stream is not None:
                            r.stream.put(None)   # release SSE readers
                        r.event.set()
                    self._q.clear()
                    for b, st in enumerate(self._slots):
                        if st is not None:
                            st["req"].error = err
                            if st["req"].stream is not None:
                                st["req"].stream.put(None)
                            st["req"].event.set()
                            self._slots[b] = None
                    return
                pending = []
                for b in range(self.max_slots):
                    if self._slots[b] is None and self._q:
                        pending.append((b, self._q.popleft()))
            try:
                self._ensure_cache()
                for b, req in pending:
                    try:
                        self._admit(b, req)
                        if self._slots[b]["count"] >= \
                                self._slots[b]["req"].max_new_tokens:
                            self._flush_stream(b, self._slots[b],
                                               final=True)
                            self._finish(b)
                    except Exception as e:   # bad request: surface it
                        req.error = e
                        req.event.set()
                if any(s is not None for s in self._slots):
                    self._step()
            except Exception as e:          # engine fault: fail occupants
                for b, st in enumerate(self._slots):
                    if st is not None:
                        st["req"].error = e
                        if st["req"].stream is not None:
                            st["req"].stream.put(None)
                        st["req"].event.set()
                        self._slots[b] = None


def create_app(model: torch.nn.Module, max_batch: int = 8,
               window_s: float = DEFAULT_WINDOW_S, use_graph: bool = False,
               tokenizer=None, continuous: bool = False,
               draft_model=None, draft_k: int = 4, worker=None):
    """FastAPI app around an inference worker.  `model` is an eval-mode
    LM from scaffold.models (tests inject a tiny CPU one).  Passing a
    `tokenizers.Tokenizer` adds the text endpoint /v1/completions.
    continuous=True serves from the ContinuousBatcher (slot-level
    admission/retirement) instead of the same-shape batch worker."""
    from contextlib import asynccontextmanager

    from fastapi import FastAPI, HTTPException

    if worker is None:
        if continuous:
            worker = ContinuousBatcher(model.eval(), max_slots=max_batch,
                                       use_graph=use_graph)
        else:
            worker = InferenceWorker(model.eval(), max_batch=max_batch,
                                     window_s=window_s, use_graph=use_graph,
                                     draft_model=draft_model,
                                     draft_k=draft_k)

    @asynccontextmanager
    async def _lifespan(app):
        yield
        worker.close()

    app = FastAPI(title="scaffold-serve", lifespan=_lifespan)
    app.state.worker = worker

    @app.get("/healthz")
    def healthz():
        return {"ok": True}

    @app.get("/v1/stats")
    def stats():
        s = dict(worker.stats)
        s["uptime_s"] = time.time() - s.pop("started_at")
        if s.get("batches"):
            s["mean_batch"] = s["requests"] / s["batches"]
        if s.get("steps"):
            # continuous batching: average decode-slot occupancy
            s["mean_occupancy"] = s["tokens_generated"] / s["steps"]
        lat = sorted(worker.latencies_ms)
        if lat:
            s["latency_ms"] = {"p50": _percentile(lat, 0.5),
                               "p95": _percentile(lat, 0.95),
                               "max": lat[-1]}
        return s

    @app.get("/metrics")
    def metrics():
        from starlette.responses import PlainTextResponse

        st_ = dict(worker.stats)
        up = time.time() - st_.pop("started_at")
        lines = [
            "# TYPE scaffold_requests_total counter",
            f"scaffold_requests_total {st_.get('requests', 0)}",
            "# TYPE scaffold_tokens_generated_total counter",
            f"scaffold_tokens_generated_total {st_.get('tokens_generated', 0)}",
            "# TYPE scaffold_uptime_seconds gauge",
            f"scaffold_uptime_seconds {up:.3f}",
        ]
        for k in ("batches", "steps", "prefix_hits",
                  "prefix_tokens_reused"):
            if k in st_:
                lines += [f"# TYPE scaffold_{k}_total counter",
                          f"scaffold_{k}_total {st_[k]}"]
        lat = sorted(worker.latencies_ms)
        if lat:
            lines += [
                "# TYPE scaffold_request_latency_ms summary",
                f'scaffold_request_latency_ms{{quantile="0.5"}} '
                f"{_percentile(lat, 0.5):.3f}",
                f'scaffold_request_latency_ms{{quantile="0.95"}} '
                f"{_percentile(lat, 0.95):.3f}",
            ]
        return PlainTextResponse("\n".join(lines) + "\n",
                                 media_type="text/plain; version=0.0.4")

    @app.post("/v1/generate")
    async def generate(body: GenRequest):
        prompts = body.tokens
        if prompts and isinstance(prompts[0], int):
            prompts = [prompts]
        if not prompts:
            raise HTTPException(400, "tokens must be a non-empty id list")
        if body.stream:
            import json
            import queue as _queue

            from starlette.responses import StreamingResponse

            if len(prompts) != 1:
                raise HTTPException(400, "stream=true takes ONE prompt")
            if not isinstance(worker, ContinuousBatcher):
                raise HTTPException(400,
                                    "streaming needs --continuous mode")
            q = _queue.Queue()
            try:
                req = worker.submit(prompts[0], body.max_new_tokens,
                                    body.temperature, body.top_k,
                                    body.top_p,
                                    eos_token_id=body.eos_token_id,
                                    stream=q)
            except ValueError as e:
                raise HTTPException(400, str(e))

            async def _events():
                loop = asyncio.get_running_loop()
                while True:
                    tok = await loop.run_in_executor(None, q.get)
                    if tok is None:
                        break
                    yield f"data: {json.dumps({'token': tok})}\n\n"
                if req.error is not None:
                    yield f"data: {json.dumps({'error': repr(req.error)})}\n\n"
                yield "data: [DONE]\n\n"

            return StreamingResponse(_events(),
                                     media_type="text/event-stream")
        t0 = time.time()
        try:
            reqs = [worker.submit(p, body.max_new_tokens, body.temperature,
                                  body.top_k, body.top_p,
                                  eos_token_id=body.eos_token_id)
                    for p in prompts]
        except OverloadedError as e:
            raise HTTPException(503, str(e))
        except ValueError as e:
            raise HTTPException(400, str(e))
        loop = asyncio.get_running_loop()
        for r in reqs:
            await loop.run_in_executor(None, r.event.wait)
        for r in reqs:
            if r.error is not None:
                raise HTTPException(500, f"generation failed: {r.error!r}")
        return {"tokens": [r.result for r in reqs],
                "latency_ms": (time.time() - t0) * 1e3}

    async def _run(reqs, t0):
        loop = asyncio.get_running_loop()
        for r in reqs:
            await loop.run_in_executor(None, r.event.wait)
        for r in reqs:
            if r.error is not None:
                raise HTTPException(500, f"generation failed: {r.error!r}")
        return (time.time() - t0) * 1e3

    if tokenizer is not None:
        @app.post("/v1/completions")
        async def completions(body: TextRequest):
            prompts = [body.prompt] if isinstance(body.prompt, str) \
                else list(body.prompt)
            if not prompts or not all(isinstance(p, str) and p
                                      for p in prompts):
                raise HTTPException(400, "prompt must be non-empty text")
            t0 = time.time()
            try:
                reqs = [worker.submit(tokenizer.encode(p).ids,
                                      body.max_new_tokens, body.temperature,
                                      body.top_k, body.top_p,
                                      eos_token_id=body.eos_token_id)
                        for p in prompts]
            except OverloadedError as e:
                raise HTTPException(503, str(e))
            except ValueError as e:
                raise HTTPException(400, str(e))
            ms = await _run(reqs, t0)
            return {"text": [tokenizer.decode(r.result) for r in reqs],
                    "tokens": [r.result for r in reqs], "latency_ms": ms}

    return app


def main(argv=None):
    import argparse

    from .config import load_config
    from .models import build_model

    ap = argparse.ArgumentParser(description=__doc__.splitlines()[0])
    ap.add_argument("overrides", nargs="*", help="config key=value overrides")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--max-batch", type=int, default=8)
    ap.add_argument("--graph", action="store_true",
                    help="hipGraph-replayed decode (fixed shapes)")
    ap.add_argument("--tokenizer", default="",
                    help="tokenizer.json path: adds text /v1/completions")
    ap.add_argument("--continuous", action="store_true",
                    help="continuous batching (per-slot admission) "
                         "instead of same-shape batch grouping")
    ap.add_argument("--weights", default="",
                    help="initial weights: safetensors file, int8 "
                         "quantized export, or HF-layout directory "
                         "(default: random init)")
    ap.add_argument("--lora", default="",
                    help="LoRA adapter state to merge into the weights "
                         "(requires --lora-r to rebuild the adapters)")
    ap.add_argument("--lora-r", type=int, default=0)
    ap.add_argument("--lora-alpha", type=float, default=16.0)
    ap.add_argument("--draft", default="",
                    help="draft model weights for speculative decoding "
                         "(safetensors; needs --draft-config overrides "
                         "as a comma-joined list, e.g. "
                         "'model.name=llama,model.n_layers=4')")
    ap.add_argument("--draft-config", default="")
    ap.add_argument("--draft-k", type=int, default=4)
    args = ap.parse_args(argv)

    import uvicorn
    cfg = load_config(None, args.overrides)
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    dt = torch.bfloat16 if dev == "cuda" else torch.float32
    model = build_model(cfg)
    if args.weights:
        from .trainer import Trainer

        Trainer._load_init_weights(model, args.weights)
    if args.lora:
        import safetensors.torch as _st

        from .lora import apply_lora, load_lora_state_dict, merge_lora

        if args.lora_r <= 0:
            ap.error("--lora needs --lora-r (the adapters' rank)")
        apply_lora(model, r=args.lora_r, alpha=args.lora_alpha)
        load_lora_state_dict(model, _st.load_file(args.lora))
        merge_lora(model)          # serve the baked weights
    model = model.to(dev, dt).eval()
    tok = None
    if args.tokenizer:
        from tokenizers import Tokenizer
        tok = Tokenizer.from_file(args.tokenizer)
    draft = None
    if args.draft:
        from .trainer import Trainer

        dcfg = load_config(None, args.draft_config.split(",")
                           if args.draft_config else [])
        draft = build_model(dcfg)
        Trainer._load_init_weights(draft, args.draft)
        draft = draft.to(dev, dt).eval()
    app = create_app(model, max_batch=args.max_batch, use_graph=args.graph,
                     tokenizer=tok, continuous=args.continuous,
                     draft_model=draft, draft_k=args.draft_k)
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
