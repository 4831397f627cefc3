"""Checkpoint layout (contract C3) — frozen behind this one module.

Layout v1:

    step_{N:08d}/
      shard_{rank}.safetensors   # this rank's owned tensors (+ rank-local RNG)
      manifest.json              # written LAST by rank 0 (atomic commit marker)

manifest.json keys: layout_version, step, world_size, config_hash,
tensors: {name: {shard, shape, dtype}}.  A checkpoint directory without
a manifest is incomplete by definition (crash mid-save) and is ignored
by resume discovery.

Ownership: model params + optimizer master/moment tensors are
partitioned across ranks greedily by size (balanced bytes per shard);
RNG/data state is rank-local and lives in every shard.  Load is
by-name from the manifest, so resume works across world-size changes
(rank-local RNG restores only on matching topology).
"""

from __future__ import annotations

import json
import os
from typing import Dict, List, Tuple

LAYOUT_VERSION = 1


def shard_file(rank: int) -> str:
    return f"shard_{rank}.safetensors"


def manifest_path(ckpt_dir: str) -> str:
    return os.path.join(ckpt_dir, "manifest.json")


def partition_by_size(names_sizes: List[Tuple[str, int]], world: int) -> Dict[str, int]:
    """Greedy balanced partition: tensor name -> owning rank."""
    loads = [0] * world
    owner: Dict[str, int] = {}
    for name, size in sorted(names_sizes, key=lambda t: -t[1]):
        r = min(range(world), key=lambda i: loads[i])
        owner[name] = r
        loads[r] += size
    return owner


def write_manifest(ckpt_dir: str, *, step: int, world_size: int, cfg_hash: str,
                   tensors: Dict[str, dict]) -> None:
    m = {
        "layout_version": LAYOUT_VERSION,
        "step": step,
        "world_size": world_size,
        "config_hash": cfg_hash,
        "tensors": tensors,
    }
    tmp = manifest_path(ckpt_dir) + ".tmp"
    with open(tmp, "w") as fh:
        json.dump(m, fh, indent=1)
        fh.flush()
        os.fsync(fh.fileno())
    os.replace(tmp, manifest_path(ckpt_dir))


def read_manifest(ckpt_dir: str) -> dict:
    p = manifest_path(ckpt_dir)
    if not os.path.exists(p):
        raise FileNotFoundError(
            f"{p} missing — incomplete checkpoint (no manifest => not committed)"
        )
    with open(p) as fh:
        m = json.load(fh)
    if m.get("layout_version") != LAYOUT_VERSION:
        raise ValueError(
            f"checkpoint layout_version {m.get('layout_version')} != {LAYOUT_VERSION}"
        )
    return m


def latest_checkpoint(base_dir: str) -> str:
    """Newest COMPLETE step_* dir (manifest present) under base_dir."""
    best = ""
    best_step = -1
    if not os.path.isdir(base_dir):
        return ""
    for d in os.listdir(base_dir):
        if d.startswith("step_") and os.path.exists(manifest_path(os.path.join(base_dir, d))):
            try:
                s = int(d.split("_")[1])
            except (IndexError, ValueError):
                continue
            if s > best_step:
                best_step = s
                best = os.path.join(base_dir, d)
    return best
