"""Trainer — the drop-in API surface (contract C1): Trainer(cfg).fit().

Step loop: synthetic batch (device-side) -> forward ->
backward (bucket all-reduces fire from grad hooks, overlapped) ->
ddp.finish() -> fused AdamW step (grad averaged via grad_scale=1/world)
-> metrics / checkpoint cadence.
"""

from __future__ import annotations

import os
import time
from typing import Optional

import torch

from .config import TrainConfig, config_hash
from .schedule import lr_at
from .data.synthetic import build_data
from .dist.comm import Comm
from .dist.ddp import BucketedDDP
from .metrics import MetricLogger, StepBreakdown, hbm_stats
from .models import build_model
from .optim import build_optimizer, MasterAdamW
from .runtime import DistContext, init_distributed, teardown


DTYPES = {"bf16": torch.bfloat16, "fp32": torch.float32}


class Trainer:
    def __init__(
        self,
        cfg: TrainConfig,
        ctx: Optional[DistContext] = None,
        model: Optional[torch.nn.Module] = None,
        data=None,
        comm: Optional[Comm] = None,
    ):
        self.cfg = cfg
        self.ctx = ctx if ctx is not None else init_distributed(cfg)
        self.dtype = DTYPES[cfg.dtype]
        self.device = self.ctx.device
        # Deterministic init regardless of how ctx was constructed; rank 0's
        # init wins via the wrap-time broadcast, so model init matches a
        # single-process run with the same seed.
        from .runtime import set_seed

        set_seed(cfg.seed, self.ctx.rank)

        if getattr(cfg.dist, "ep_moe", False) and self.ctx.world_size > 1:
            if getattr(cfg.dist, "zero1", False):
                raise ValueError("dist.ep_moe and dist.zero1 are mutually "
                                 "exclusive (expert params are rank-local)")
            # expert-parallel MoE: build_model shards experts by rank.
            # The routed-token all_to_alls run during BACKWARD and would
            # interleave with the bucket all-reduces in a readiness-
            # dependent (rank-INconsistent) order — a collective
            # mismatch on a single communicator (gloo detects it; RCCL
            # would corrupt).  A dedicated EP group gives the a2a its
            # own communicator, making the two streams order-independent.
            import torch.distributed as dist

            ep_group = dist.new_group(list(range(self.ctx.world_size)))
            cfg._ep_ctx = (self.ctx.world_size, self.ctx.rank, ep_group)
        raw_model = model if model is not None else build_model(cfg)
        raw_model = raw_model.to(self.device)
        if self.dtype is not torch.float32:
            raw_model = self._to_mixed(raw_model)
        self.channels_last = cfg.model.name == "resnet50" and self.ctx.on_gpu
        if self.ctx.on_gpu:
            # MIOpen: exhaustive algo find; results persist in the
            # in-tree miopen_db so later runs skip the search.  Set
            # cudnn_benchmark=false for heuristic (immediate) algo
            # choice when the search cost cannot be amortized.
            torch.backends.cudnn.benchmark = bool(cfg.cudnn_benchmark)
        if self.channels_last:
            # NHWC is MIOpen's fast conv layout on CDNA
            raw_model = raw_model.to(memory_format=torch.channels_last)
        if cfg.ckpt.init_weights:
            # fine-tune initialization (BEFORE LoRA freeze and before
            # the DDP wrap-time broadcast, which then distributes it)
            self._load_init_weights(raw_model, cfg.ckpt.init_weights)
        if getattr(cfg.model, "lora_r", 0) > 0:
            from .lora import apply_lora

            apply_lora(raw_model, r=cfg.model.lora_r,
                       alpha=cfg.model.lora_alpha)
        self.raw_model = raw_model

        from .dist.comm import make_comm

        self.ddp = BucketedDDP(
            raw_model,
            comm=comm if comm is not None else make_comm(cfg, self.ctx),
            bucket_cap_mb=cfg.dist.bucket_cap_mb,
            first_bucket_mb=cfg.dist.first_bucket_mb,
            overlap=cfg.dist.overlap,
        )
        self.model = self.ddp
        self.optimizer: MasterAdamW = build_optimizer(cfg, raw_model, self.ctx)
        # ZeRO-1 (dist.zero1): after the sharded optimizer step, owners
        # push their updated working params to the group.  Broadcasts
        # are COALESCED: one flat buffer per (owner, dtype) group
        # (Llama-8B has ~300 params — 8 buffer broadcasts instead of
        # 300 small ones on RCCL), packed by the owner, scattered back
        # by everyone else.
        self._zero1_groups = None
        if self.optimizer.shard_owners is not None:
            groups = {}
            for n, p in raw_model.named_parameters():
                if not p.requires_grad:
                    continue
                key = (self.optimizer.shard_owners[n], p.dtype)
                groups.setdefault(key, []).append(p)
            self._zero1_groups = []
            for (owner, dtype), ps in sorted(
                    groups.items(), key=lambda kv: (kv[0][0], str(kv[0][1]))):
                flat = torch.empty(sum(p.numel() for p in ps), dtype=dtype,
                                   device=self.device)
                self._zero1_groups.append((owner, ps, flat))
        self.data = data if data is not None else build_data(
            cfg, self.ctx.rank, self.device, self.dtype
        )
        self.logger = MetricLogger(self.ctx, cfg)
        self.breakdown = StepBreakdown(self.device)
        self.step = 0
        self.cfg_hash = config_hash(cfg)
        if cfg.ckpt.resume:
            self._resume(cfg.ckpt.resume)

    @staticmethod
    def _load_init_weights(model: torch.nn.Module, path: str) -> None:
        """Dispatch on the artifact kind: HF directory, int8 quantized
        export, or plain safetensors weights."""
        import os as _os

        from . import convert

        if _os.path.isdir(path):
            convert.load_hf_dir(model, path)
            return
        from safetensors import safe_open

        with safe_open(path, framework="pt", device="cpu") as f:
            quantized = any(k.endswith("::int8") for k in f.keys())
        if quantized:
            from .quant import load_quantized

            load_quantized(model, path)
        else:
            convert.load_model(model, path)

    def _to_mixed(self, model: torch.nn.Module) -> torch.nn.Module:
        """bf16 working weights; BatchNorm stays fp32 (numerics)."""
        model = model.to(self.dtype)
        for mod in model.modules():
            if isinstance(mod, (torch.nn.BatchNorm1d, torch.nn.BatchNorm2d)):
                mod.float()
        return model

    # --------------------------------------------------------------- fit
    def fit(self, steps: Optional[int] = None):
        total = steps if steps is not None else self.cfg.steps
        accum = max(1, self.cfg.grad_accum)
        batch_iter = iter(self.data)
        prof = None
        if self.cfg.log.profile_steps > 0:
            # torch.profiler works on ROCm; kernel-level
            # counters come from rocprofv3 runs instead (profiles/).
            prof = torch.profiler.profile(
                activities=[torch.profiler.ProfilerActivity.CPU,
                            torch.profiler.ProfilerActivity.CUDA],
                schedule=torch.profiler.schedule(
                    wait=2, warmup=2, active=self.cfg.log.profile_steps),
                on_trace_ready=lambda p: p.export_chrome_trace(
                    f"trace_rank{self.ctx.rank}.json"),
            )
            prof.__enter__()
        use_graph = self._graph_ok()
        if use_graph and not hasattr(self, "_graph"):
            try:
                self._capture_step(batch_iter)
            except Exception as e:  # noqa: BLE001 — e.g. non-capturable RNG
                print(f"[scaffold] hipGraph capture failed ({e}); eager fallback",
                      flush=True)
                self.optimizer.step_dev = None
                use_graph = False
        # test-only fault injection for the elastic-launcher tier
        # (SURVEY §5.3): SCAFFOLD_FAULT_STEP/_RANK kill this rank
        # mid-run, uncleanly, exactly once (the relaunched world clears
        # the guard via SCAFFOLD_FAULT_DONE in its env).
        fault_step = int(os.environ.get("SCAFFOLD_FAULT_STEP", "-1"))
        fault_rank = int(os.environ.get("SCAFFOLD_FAULT_RANK", "0"))
        while self.step < total:
            if (fault_step >= 0 and self.step == fault_step
                    and self.ctx.rank == fault_rank
                    and "SCAFFOLD_FAULT_DONE" not in os.environ):
                os._exit(17)
            t0 = time.perf_counter()
            loss_val = (self._train_step_graph() if use_graph
                        else self._train_step(batch_iter, accum))
            dt = time.perf_counter() - t0
            self.step += 1
            extra = None
            if self.step % max(1, self.cfg.log.interval) == 0:
                # measured breakdowns drain events — log cadence only,
                # never per step (§5.5)
                extra = {}
                if hasattr(self.ddp.comm, "comm_time_ms"):
                    # per-bucket collective time on the comm stream
                    extra["comm_ms"] = round(self.ddp.comm.comm_time_ms(), 3)
                extra.update(self.breakdown.drain())
                extra.update(hbm_stats(self.device))
            self.logger.log_step(
                self.step, loss_val, self.cfg.data.batch_size * accum, dt,
                extra=extra,
            )
            if prof is not None:
                prof.step()
            if (self.cfg.log.eval_interval
                    and self.step % self.cfg.log.eval_interval == 0):
                ev = self.evaluate(self.cfg.log.eval_steps)
                if self.ctx.is_main:
                    print(f"step {self.step:6d} | eval loss {ev:8.4f}", flush=True)
            if (
                self.cfg.ckpt.interval
                and self.step % self.cfg.ckpt.interval == 0
                and self.cfg.ckpt.dir
            ):
                self.save_checkpoint()
        if prof is not None:
            prof.__exit__(None, None, None)
        if self.cfg.ckpt.save_final and self.cfg.ckpt.dir:
            self.save_checkpoint()
        return self

    # ------------------------------------------------------ hipGraph path
    def _graph_ok(self) -> bool:
        """hipGraph replay freezes launch-time scalars, so capture is
        only valid when nothing host-side varies per step: constant lr,
        no grad clip (host .item()), no grad accumulation branching."""
        return (self.ctx.on_gpu and self.cfg.hip_graph
                and self.cfg.optim.schedule == "const"
                and self.cfg.optim.warmup_steps == 0
                and self.cfg.optim.grad_clip == 0
                and self.cfg.grad_accum <= 1
                and self._zero1_groups is None)  # collectives in step

    def _capture_step(self, batch_iter):
        """Capture ONE whole training step (synthetic-batch generation,
        forward, backward with bucket all-reduces, fused AdamW) into a
        hipGraph (torch.cuda.CUDAGraph is hipGraph on ROCm).  Synthetic
        data is generated in-graph: torch replays RNG with graph-safe
        philox state, so every replay sees fresh data.  ~3-4 us of host
        launch per kernel collapses to one graph launch (MI355X_MICROARCH
        'graph-replay-floor')."""
        # warmup on a side stream (allocator pools + MIOpen/algo choices)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._train_step(batch_iter, 1)
        torch.cuda.current_stream().wait_stream(s)

        # device-side step counter: incremented IN-GRAPH so the AdamW
        # bias corrections stay exact under replay (adamw.hip step_dev)
        self.optimizer.step_dev = torch.tensor(
            [self.optimizer.step_count], dtype=torch.int32, device=self.device)

        g = torch.cuda.CUDAGraph()
        # relaxed: MIOpen issues stream queries during capture that the
        # default "global" mode rejects (hipErrorStreamCaptureUnsupported
        # on the ResNet config); the step itself is capture-safe
        with torch.cuda.graph(g, capture_error_mode="relaxed"):
            self.ddp.zero_grad()
            x, y = self.data.next_batch()
            loss = self.model(x, y)
            loss.backward()
            self.ddp.finish()
            self.optimizer.step_dev.add_(1)
            self.optimizer.step(grad_scale=self.ddp.grad_scale)
        self._graph = g
        self._graph_loss = loss
        # capture RECORDS without executing, but the host-side
        # step_count bump inside optimizer.step ran — cancel it
        self.optimizer.step_count -= 1

    def _train_step_graph(self) -> float:
        self._graph.replay()
        self.optimizer.step_count += 1  # host mirror (checkpoints)
        return float(self._graph_loss.detach().float().item())

    def _train_step(self, batch_iter, accum: int) -> float:
        bd = self.breakdown
        bd.start_step()
        self.ddp.zero_grad()
        loss_val = 0.0
        for micro in range(accum):
            x, y = next(batch_iter)
            bd.mark("data")
            if micro < accum - 1:
                with self.ddp.no_sync():
                    loss = self.model(x, y)
                    bd.mark("fwd")
                    loss.backward()
            else:
                loss = self.model(x, y)
                bd.mark("fwd")
                loss.backward()
                self.ddp.finish()
            bd.mark("bwd")
            loss_val += loss.detach().float().item() / accum

        self.optimizer.lr = lr_at(self.cfg, self.step, self.cfg.steps)
        scale = self.ddp.grad_scale / accum
        if self.cfg.optim.grad_clip > 0:
            # global-norm clip FOLDED into the fused step's grad_scale:
            # norm of the averaged grads = ||flat buckets|| * scale, so
            # the clip costs a few _foreach_norm launches and ZERO extra
            # full-gradient passes.
            norms = torch._foreach_norm([b.flat for b in self.ddp.buckets])
            gnorm = float(torch.linalg.vector_norm(torch.stack(norms))) * scale
            if gnorm > self.cfg.optim.grad_clip:
                scale *= self.cfg.optim.grad_clip / (gnorm + 1e-12)
        self.optimizer.step(grad_scale=scale)
        if self._zero1_groups is not None:
            import torch.distributed as dist

            works = []
            for owner, ps, flat in self._zero1_groups:
                if self.ctx.rank == owner:
                    off = 0
                    for p in ps:
                        flat[off:off + p.numel()].copy_(p.data.flatten())
                        off += p.numel()
                works.append(dist.broadcast(flat, src=owner, async_op=True))
            for (owner, ps, flat), w in zip(self._zero1_groups, works):
                w.wait()
                if self.ctx.rank != owner:
                    off = 0
                    for p in ps:
                        p.data.copy_(
                            flat[off:off + p.numel()].view_as(p.data))
                        off += p.numel()
        bd.mark("opt")
        bd.end_step()
        return loss_val

    # -------------------------------------------------------------- eval
    @torch.no_grad()
    def evaluate(self, eval_steps: int = 10) -> float:
        self.raw_model.eval()
        total = 0.0
        # Eval must not perturb the training data stream: snapshot the
        # generator state and restore it after drawing eval batches, so
        # training with eval cadence on sees the same data as with it off.
        saved_gen = self.data.state_dict()
        it = iter(self.data)
        for _ in range(eval_steps):
            x, y = next(it)
            total += self.model.module(x, y).float().item()
        self.data.load_state_dict(saved_gen)
        self.raw_model.train()
        # Reduction tensor lives on the PG's device: the GPU process
        # group is RCCL ("nccl"), which cannot reduce CPU tensors.
        t = torch.tensor([total / eval_steps], device=self.device)
        self.ddp.comm.all_reduce(t)
        return t.item() / max(1, self.ddp.comm.world_size)

    # --------------------------------------------------------- checkpoint
    def save_checkpoint(self, path: Optional[str] = None) -> str:
        from .ckpt.sharded import save_sharded

        out = path or os.path.join(self.cfg.ckpt.dir, f"step_{self.step:08d}")
        save_sharded(
            out,
            step=self.step,
            model=self.raw_model,
            optimizer=self.optimizer,
            data=self.data,
            ctx=self.ctx,
            cfg_hash=self.cfg_hash,
            async_d2h=self.cfg.ckpt.async_d2h,
        )
        return out

    def _resume(self, path: str) -> None:
        from .ckpt.sharded import load_sharded

        self.step = load_sharded(
            path,
            model=self.raw_model,
            optimizer=self.optimizer,
            data=self.data,
            ctx=self.ctx,
            cfg_hash=self.cfg_hash,
        )

    def close(self) -> None:
        self.logger.close()
        teardown(self.ctx)
