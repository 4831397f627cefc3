"""Sharded checkpoint save/load (contracts C3, C6, C12).

Save path on GPU: device tensors stream to host through PINNED staging
buffers with async D2H copies (hipMemcpyAsync under torch's
``Tensor.copy_(non_blocking=True)``) on a DEDICATED side stream, so the
compute stream is never blocked (C6).  Chunks are event-signalled; the
host thread serializes a chunk to the shard file only after its event
fires.  On CPU the same code degenerates to plain copies.

Format: safetensors per shard + manifest.json (layout.py).

Load is streaming same-world (each rank reads only its own shard from
disk, the full state completes over collectives through a pinned H2D
double-buffer) or by-name resharding across world sizes.  ZeRO-1
optimizers (optimizer.shard_owners set) save their state RANK-PINNED:
each rank's shard carries its owned masters/moments, the manifest stays
globally consistent because the ownership partition is deterministic,
and the same-world loader reads them locally without collectives.
"""

from __future__ import annotations

import os
from typing import Dict, List, Tuple

import torch
from safetensors.torch import save_file, safe_open

from . import layout


def _named_state_tensors(model, optimizer) -> List[Tuple[str, torch.Tensor]]:
    """Flat name->tensor list: model params+buffers, optimizer masters/moments."""
    out: List[Tuple[str, torch.Tensor]] = []
    for name, p in model.named_parameters():
        # rank-pinned params (EP experts) carry global identities
        out.append((f"model.{getattr(p, '_ckpt_name', name)}", p.data))
    for name, b in model.named_buffers():
        out.append((f"buffer.{name}", b.data))
    if optimizer is not None:
        # ZeRO-1 subset optimizers carry their own name identities
        pnames = (optimizer.param_names
                  if getattr(optimizer, "param_names", None) is not None
                  else [n for n, _ in model.named_parameters()])
        for i, t in enumerate(optimizer.master):
            out.append((f"opt.master.{pnames[i]}", t))
        for i, t in enumerate(optimizer.exp_avg):
            out.append((f"opt.exp_avg.{pnames[i]}", t))
        for i, t in enumerate(optimizer.exp_avg_sq):
            out.append((f"opt.exp_avg_sq.{pnames[i]}", t))
    return out


class _PinnedStager:
    """Double-buffered pinned ring: D2H on a side stream, event-gated reads."""

    def __init__(self, device: torch.device, nbuf: int = 4, buf_bytes: int = 64 << 20):
        self.on_gpu = device.type == "cuda"
        if self.on_gpu:
            self.stream = torch.cuda.Stream(device)
            self.bufs = [
                torch.empty(buf_bytes, dtype=torch.uint8, pin_memory=True)
                for _ in range(nbuf)
            ]
            self.events = [torch.cuda.Event() for _ in range(nbuf)]
            self.buf_bytes = buf_bytes
            self.slot = 0

    def to_host(self, t: torch.Tensor) -> torch.Tensor:
        """Return a CPU copy of t; async through the pinned ring on GPU."""
        if not self.on_gpu or not t.is_cuda:
            return t.detach().cpu()
        # order after in-flight compute (e.g. the optimizer step that
        # produced this state) before copying on the side stream
        self.stream.wait_stream(torch.cuda.current_stream())
        flat = t.detach().contiguous().view(torch.uint8).view(-1)
        n = flat.numel()
        if n > self.buf_bytes:
            # big tensor: one dedicated pinned buffer, still on side stream
            host = torch.empty(n, dtype=torch.uint8, pin_memory=True)
            with torch.cuda.stream(self.stream):
                host.copy_(flat, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record(self.stream)
            ev.synchronize()
            return host.view(t.dtype).view(t.shape).clone()
        s = self.slot
        self.slot = (s + 1) % len(self.bufs)
        self.events[s].synchronize()  # previous use of this slot done
        buf = self.bufs[s][:n]
        with torch.cuda.stream(self.stream):
            buf.copy_(flat, non_blocking=True)
        self.events[s].record(self.stream)
        self.events[s].synchronize()
        return buf.view(t.dtype).view(t.shape).clone()


def save_sharded(ckpt_dir: str, *, step: int, model, optimizer, data, ctx,
                 cfg_hash: str, async_d2h: bool = True) -> None:
    os.makedirs(ckpt_dir, exist_ok=True)
    world = ctx.world_size
    rank = ctx.rank

    named = _named_state_tensors(model, optimizer)
    # ZeRO-1: optimizer state is RANK-PINNED (each rank owns a shard of
    # masters/moments by construction), so it bypasses the size-balanced
    # partition; every rank can still build the full manifest because
    # the ownership map is deterministic and masters mirror param shapes
    zero1 = getattr(optimizer, "shard_owners", None) is not None
    pinned = []
    if zero1:
        pinned = [(n, t) for n, t in named if n.startswith("opt.")]
        named = [(n, t) for n, t in named if not n.startswith("opt.")]
    # EP experts: rank-LOCAL params (and their optimizer state) keyed
    # by global expert id — pin them to this rank; the manifest covers
    # every rank's experts via the model's _pinned_manifest map
    ep_map = getattr(model, "_pinned_manifest", None)
    if ep_map:
        ep_names = set(ep_map)

        def _pin_base(n):
            base = n.split(".", 1)[1] if n.startswith("model.") else \
                n.split(".", 2)[2] if n.startswith("opt.") else None
            return base if base in ep_names else None

        more = [(n, t) for n, t in named if _pin_base(n)]
        pinned += more
        named = [(n, t) for n, t in named if not _pin_base(n)]
    # dedupe aliased storages (fp32 params alias their optimizer master;
    # tied embeddings alias lm_head) — safetensors forbids shared tensors
    seen_ptr: Dict[int, str] = {}
    unique, aliases = [], {}
    for name, t in named:
        ptr = t.untyped_storage().data_ptr()
        if ptr in seen_ptr and t.numel() == t.untyped_storage().nbytes() // max(1, t.element_size()):
            aliases[name] = seen_ptr[ptr]
        else:
            seen_ptr.setdefault(ptr, name)
            unique.append((name, t))
    owner = layout.partition_by_size(
        [(n, t.numel() * t.element_size()) for n, t in unique], world
    )

    shard: Dict[str, torch.Tensor] = {}
    tensor_meta: Dict[str, dict] = {}
    for name, t in unique:
        tensor_meta[name] = {
            "shard": owner[name],
            "shape": list(t.shape),
            "dtype": str(t.dtype).replace("torch.", ""),
        }
        if owner[name] == rank:
            shard[name] = t.detach()
    for name, target in aliases.items():
        tensor_meta[name] = {"alias": target}
    if zero1:
        shapes = {n: list(p.shape) for n, p in model.named_parameters()}
        for pname, r in optimizer.shard_owners.items():
            for kind in ("master", "exp_avg", "exp_avg_sq"):
                tensor_meta[f"opt.{kind}.{pname}"] = {
                    "shard": r, "shape": shapes[pname], "dtype": "float32"}
    if ep_map:
        pdtype = str(next(model.parameters()).dtype).replace("torch.", "")
        for base, (owner_r, shape) in ep_map.items():
            tensor_meta[f"model.{base}"] = {
                "shard": owner_r, "shape": shape, "dtype": pdtype}
            if optimizer is not None:
                for kind in ("master", "exp_avg", "exp_avg_sq"):
                    tensor_meta[f"opt.{kind}.{base}"] = {
                        "shard": owner_r, "shape": shape,
                        "dtype": "float32"}
    if pinned:
        taken = {t.untyped_storage().data_ptr() for t in shard.values()}
        for name, t in pinned:
            t = t.detach()
            # fp32 masters alias the working param (already in this
            # shard when the save-partition assigned it here): clone so
            # safetensors never sees shared storage
            if t.untyped_storage().data_ptr() in taken:
                t = t.clone()
            shard[name] = t
            taken.add(t.untyped_storage().data_ptr())

    # rank-local RNG + data state (every shard)
    shard[f"rng.torch.{rank}"] = torch.random.get_rng_state()
    if ctx.on_gpu:
        shard[f"rng.cuda.{rank}"] = torch.cuda.get_rng_state(ctx.device)
    if data is not None:
        shard[f"rng.data.{rank}"] = data.state_dict()["gen_state"].cpu()
    sf = os.path.join(ckpt_dir, layout.shard_file(rank))

    from ..ops.backend import ext as _ext

    native = _ext() if (ctx.on_gpu and async_d2h) else None
    if native is not None:
        # C6 native path: device tensors stream straight to the shard
        # file through the pinned double-buffer on a side stream
        # (csrc/io/pinned_stager.hip); file is safetensors-compatible.
        names = list(shard.keys())
        native.save_shard_pinned(
            sf, names, [shard[n].contiguous() for n in names],
            ["rank", "step"], [str(rank), str(step)],
        )
    else:
        stager = _PinnedStager(ctx.device) if async_d2h else None
        host_shard = {}
        for name, t in shard.items():
            host = stager.to_host(t) if (stager and t.is_cuda) \
                else t.detach().contiguous().cpu()
            host_shard[name] = host.contiguous()
        save_file(host_shard, sf, metadata={"rank": str(rank), "step": str(step)})
        with open(sf, "rb") as fh:
            os.fsync(fh.fileno())

    # all shards durable before rank 0 commits the manifest
    if ctx.initialized:
        import torch.distributed as dist

        dist.barrier()
    if rank == 0:
        meta = dict(tensor_meta)
        meta["opt.step"] = {"shard": -1, "value": optimizer.step_count if optimizer else 0}
        layout.write_manifest(
            ckpt_dir, step=step, world_size=world, cfg_hash=cfg_hash, tensors=meta
        )
    if ctx.initialized:
        import torch.distributed as dist

        dist.barrier()


def _dest_tensors(model, optimizer, manifest) -> List[Tuple[str, torch.Tensor]]:
    """name -> destination tensor list, mirroring _named_state_tensors."""
    out: List[Tuple[str, torch.Tensor]] = []
    pnames = []
    for name, p in model.named_parameters():
        name = getattr(p, "_ckpt_name", name)   # EP experts: global ids
        pnames.append(name)
        out.append((f"model.{name}", p.data))
    for name, b in model.named_buffers():
        key = f"buffer.{name}"
        if key in manifest["tensors"]:
            out.append((key, b.data))
    if optimizer is not None:
        onames = (optimizer.param_names
                  if getattr(optimizer, "param_names", None) is not None
                  else pnames)
        for i, name in enumerate(onames):
            out.append((f"opt.master.{name}", optimizer.master[i]))
            out.append((f"opt.exp_avg.{name}", optimizer.exp_avg[i]))
            out.append((f"opt.exp_avg_sq.{name}", optimizer.exp_avg_sq[i]))
    return out


class _PinnedH2D:
    """Small pinned double-buffer for async host->device loads: the
    CPU->pinned memcpy of tensor n overlaps the H2D DMA of tensor n-1."""

    def __init__(self, device: torch.device, buf_bytes: int = 64 << 20):
        self.on_gpu = device.type == "cuda"
        if self.on_gpu:
            self.bufs = [torch.empty(buf_bytes, dtype=torch.uint8, pin_memory=True)
                         for _ in range(2)]
            self.events = [torch.cuda.Event() for _ in range(2)]
            self.buf_bytes = buf_bytes
            self.slot = 0

    def copy_(self, dst: torch.Tensor, src_cpu: torch.Tensor) -> None:
        if not self.on_gpu or not dst.is_cuda:
            dst.copy_(src_cpu)
            return
        src = src_cpu.to(dst.dtype).contiguous().view(torch.uint8).view(-1)
        n = src.numel()
        if n > self.buf_bytes:
            dst.copy_(src_cpu.to(dst.dtype), non_blocking=False)
            return
        s = self.slot
        self.slot = s ^ 1
        self.events[s].synchronize()  # previous H2D from this slot done
        buf = self.bufs[s][:n]
        buf.copy_(src)  # host memcpy into pinned
        dst.view(torch.uint8).view(-1).copy_(buf, non_blocking=True)
        self.events[s].record()


def load_sharded(ckpt_dir: str, *, model, optimizer, data, ctx, cfg_hash: str = "",
                 strict_hash: bool = False) -> int:
    m = layout.read_manifest(ckpt_dir)
    if strict_hash and cfg_hash and m["config_hash"] != cfg_hash:
        raise ValueError(
            f"config hash mismatch: ckpt {m['config_hash']} vs current {cfg_hash}"
        )
    saved_world = m["world_size"]

    def resolve(name: str) -> str:
        if name not in m["tensors"] and ".moe." in name:
            # reshard across EP worlds: a local-expert model asks for
            # slot index i; an EP checkpoint stored global ids gN.
            # With world=1 (or replicated experts) slot i IS global i.
            head, idx = name.rsplit(".", 1)
            alt = f"{head}.g{idx}"
            if alt in m["tensors"]:
                name = alt
        info = m["tensors"][name]
        return info["alias"] if "alias" in info else name

    # ---- streaming same-world path (SURVEY §3.5 load half of C6):
    # each rank reads ONLY its own shard from disk (1x aggregate read
    # instead of world x), stages it host->device through a pinned
    # double-buffer, and the full state is completed by per-tensor
    # broadcasts from the owning rank over RCCL/xGMI — which is orders
    # of magnitude faster than disk.
    if ctx.initialized and ctx.world_size == saved_world and saved_world > 1:
        import torch.distributed as dist

        own = safe_open(os.path.join(ckpt_dir, layout.shard_file(ctx.rank)),
                        framework="pt", device="cpu")
        stager = _PinnedH2D(ctx.device)
        with torch.no_grad():
            # one pass: owner stages its tensor host->device (async on
            # the current stream), then the broadcast — stream-ordered
            # after the copy — ships it to the other ranks; the next
            # tensor's disk read + pinned memcpy overlap both.
            # Dedup by storage (fp32 params alias their master) — the
            # alias structure is identical on every rank, so the
            # broadcast sequences stay aligned.
            zero1 = getattr(optimizer, "shard_owners", None) is not None
            ep_bases = set(getattr(model, "_pinned_manifest", {}) or {})

            def _rank_pinned(n):
                if zero1 and n.startswith("opt.") and not ep_bases:
                    return True
                base = (n.split(".", 1)[1] if n.startswith("model.")
                        else n.split(".", 2)[2] if n.startswith("opt.")
                        else None)
                return (zero1 and n.startswith("opt.")) or base in ep_bases

            seen_storage = set()
            for name, dst in _dest_tensors(model, optimizer, m):
                rname = resolve(name)
                owner = m["tensors"][rname]["shard"]
                ptr = dst.untyped_storage().data_ptr()
                if ptr in seen_storage:
                    continue
                seen_storage.add(ptr)
                if _rank_pinned(name):
                    # rank-pinned state (zero1 opt shards, EP experts):
                    # same world => it sits in OUR shard; other ranks
                    # don't even have a destination, so no collective
                    # may be issued for it
                    stager.copy_(dst, own.get_tensor(rname))
                    continue
                if owner == ctx.rank:
                    stager.copy_(dst, own.get_tensor(rname))
                dist.broadcast(dst, src=owner)
        if optimizer is not None:
            optimizer.step_count = int(
                m["tensors"].get("opt.step", {}).get("value", m["step"]))
            with torch.no_grad():
                for p, mt in zip(optimizer.params, optimizer.master):
                    p.data.copy_(mt.to(p.dtype))
        _restore_local_rng(m, {ctx.rank: own}, saved_world, ctx, data)
        return int(m["step"])

    # ---- fallback: open every shard; load-by-name (reshard-capable)
    handles = {}
    for r in range(saved_world):
        p = os.path.join(ckpt_dir, layout.shard_file(r))
        handles[r] = safe_open(p, framework="pt", device="cpu")

    def fetch(name: str) -> torch.Tensor:
        return handles[m["tensors"][resolve(name)]["shard"]].get_tensor(resolve(name))

    with torch.no_grad():
        pnames = []
        for name, p in model.named_parameters():
            pnames.append(name)
            p.data.copy_(fetch(f"model.{name}").to(p.device, p.dtype))
        for name, b in model.named_buffers():
            key = f"buffer.{name}"
            if key in m["tensors"]:
                b.data.copy_(fetch(key).to(b.device, b.dtype))
        if optimizer is not None:
            onames = (optimizer.param_names
                      if getattr(optimizer, "param_names", None) is not None
                      else pnames)
            for i, name in enumerate(onames):
                optimizer.master[i].copy_(fetch(f"opt.master.{name}").to(optimizer.master[i].device))
                optimizer.exp_avg[i].copy_(fetch(f"opt.exp_avg.{name}").to(optimizer.exp_avg[i].device))
                optimizer.exp_avg_sq[i].copy_(fetch(f"opt.exp_avg_sq.{name}").to(optimizer.exp_avg_sq[i].device))
            optimizer.step_count = int(m["tensors"].get("opt.step", {}).get("value", m["step"]))
            # working params refresh from masters
            for p, mt in zip(optimizer.params, optimizer.master):
                p.data.copy_(mt.to(p.dtype))

    _restore_local_rng(m, handles, saved_world, ctx, data)
    return int(m["step"])


def _restore_local_rng(m, handles, saved_world, ctx, data) -> None:
    """Rank-local RNG: same-topology fast path only."""
    if saved_world == ctx.world_size:
        h = handles[ctx.rank]
        keys = set(h.keys())
        if f"rng.torch.{ctx.rank}" in keys:
            torch.random.set_rng_state(h.get_tensor(f"rng.torch.{ctx.rank}"))
        if ctx.on_gpu and f"rng.cuda.{ctx.rank}" in keys:
            torch.cuda.set_rng_state(h.get_tensor(f"rng.cuda.{ctx.rank}"), ctx.device)
        if data is not None and f"rng.data.{ctx.rank}" in keys:
            data.load_state_dict({"gen_state": h.get_tensor(f"rng.data.{ctx.rank}")})
    elif data is not None:
        data.reset(step=m["step"])
