"""HTTP serving front end for the scaffold's LM families.

A single `InferenceWorker` thread owns the model and drains a request
queue in dynamic batches: pending requests with the same prompt length
and temperature are stacked into one `Generator.generate` call (the
KV-cache path, decode GEMVs batched across requests), then each
response is sliced back out.  Grouping by prompt length keeps the
batch un-padded — the Generator's cache layout has no padding mask, so
mixing lengths would silently change positions (SURVEY §5.7 seam note
applies to serving too: the op boundary stays (B, S) dense).

MI355X notes: serving decode is weight-read-bound (NOTES.md r2 serving
ladder), so batching B requests into one decode sweep multiplies
tokens/s by ~B until the batch GEMMs leave the skinny-M regime.  For
fixed-shape deployments `use_graph=True` replays the hipGraph-captured
decode step (2.5x on gpt2-small at B=1).

Endpoints:
  POST /v1/generate   {"tokens": [[int]] | [int], "max_new_tokens": N,
                       "temperature": T}  ->  {"tokens": [[int]], ...}
  GET  /healthz       liveness
  GET  /v1/stats      batching + throughput counters

Run: `python -m scaffold.serve model.name=llama ... [--port 8000]`
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

    class TextRequest(_BaseModel):
        prompt: object             # str or [str]
        max_new_tokens: int = 32
        temperature: float = 0.0
        top_k: int = 0
        top_p: float = 0.0
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
    future: "asyncio.Future | None" = None
    event: threading.Event = field(default_factory=threading.Event)
    result: Optional[List[int]] = None
    error: Optional[Exception] = None

    @property
    def key(self):
        return (int(self.tokens.numel()), round(float(self.temperature), 6),
                self.top_k, round(float(self.top_p), 6))


class InferenceWorker:
    """Owns the model; one thread serializes all GPU work.

    submit() is thread-safe and returns the _Request whose `event`
    fires when `result` (list of generated token ids, prompt excluded)
    or `error` is set.
    """

    def __init__(self, model: torch.nn.Module, max_batch: int = 8,
                 window_s: float = DEFAULT_WINDOW_S, use_graph: bool = False,
                 max_new_tokens_cap: int = 512):
        self.gen = Generator(model)
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
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="scaffold-serve-worker")
        self._thread.start()

    def submit(self, tokens: List[int], max_new_tokens: int,
               temperature: float = 0.0, top_k: int = 0,
               top_p: float = 0.0) -> _Request:
        if not tokens:
            raise ValueError("empty prompt")
        n = min(int(max_new_tokens), self.max_new_tokens_cap)
        if n < 1:
            raise ValueError("max_new_tokens must be >= 1")
        req = _Request(torch.tensor(tokens, dtype=torch.int64), n,
                       float(temperature), int(top_k), float(top_p))
        with self._cv:
            self._q.append(req)
            self._cv.notify()
        return req

    def close(self):
        with self._cv:
            self._stop = True
            self._cv.notify()
        self._thread.join(timeout=5)

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
        if self.use_graph and temp == 0.0:
            out = self.gen.generate_graphed(prompt, steps)
        else:
            out = self.gen.generate(prompt, steps, temperature=temp,
                                    top_k=batch[0].top_k,
                                    top_p=batch[0].top_p)
        s0 = prompt.shape[1]
        new = out[:, s0:].cpu()
        for i, r in enumerate(batch):
            r.result = new[i, :r.max_new_tokens].tolist()
            self.stats["tokens_generated"] += r.max_new_tokens
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


def create_app(model: torch.nn.Module, max_batch: int = 8,
               window_s: float = DEFAULT_WINDOW_S, use_graph: bool = False,
               tokenizer=None):
    """FastAPI app around an InferenceWorker.  `model` is an eval-mode
    LM from scaffold.models (tests inject a tiny CPU one).  Passing a
    `tokenizers.Tokenizer` adds the text endpoint /v1/completions."""
    from contextlib import asynccontextmanager

    from fastapi import FastAPI, HTTPException

    worker = InferenceWorker(model.eval(), max_batch=max_batch,
                             window_s=window_s, use_graph=use_graph)

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
        s["mean_batch"] = (s["requests"] / s["batches"]) if s["batches"] else 0.0
        return s

    @app.post("/v1/generate")
    async def generate(body: GenRequest):
        prompts = body.tokens
        if prompts and isinstance(prompts[0], int):
            prompts = [prompts]
        if not prompts:
            raise HTTPException(400, "tokens must be a non-empty id list")
        t0 = time.time()
        try:
            reqs = [worker.submit(p, body.max_new_tokens, body.temperature,
                                  body.top_k, body.top_p)
                    for p in prompts]
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
                                      body.top_k, body.top_p)
                        for p in prompts]
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
    args = ap.parse_args(argv)

    import uvicorn
    cfg = load_config(None, args.overrides)
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    dt = torch.bfloat16 if dev == "cuda" else torch.float32
    model = build_model(cfg).to(dev, dt).eval()
    tok = None
    if args.tokenizer:
        from tokenizers import Tokenizer
        tok = Tokenizer.from_file(args.tokenizer)
    app = create_app(model, max_batch=args.max_batch, use_graph=args.graph,
                     tokenizer=tok)
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
