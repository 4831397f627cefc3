"""Synthetic data generators (contract configs use synthetic data only).

Device-side generation: batches are produced directly on the training
device from a per-rank seeded generator, so the data path costs no host
PCIe traffic and no dataloader workers — the right design for a bench
whose data is synthetic by contract (BASELINE.json configs; no network
for datasets in this environment).
"""

from __future__ import annotations

from typing import Iterator, Tuple

import torch

from ..config import TrainConfig


class SyntheticBase:
    """Infinite per-rank deterministic batch stream."""

    def __init__(self, cfg: TrainConfig, rank: int, device: torch.device, dtype: torch.dtype):
        self.cfg = cfg
        self.rank = rank
        self.device = device
        self.dtype = dtype
        if device.type == "cuda":
            # the DEFAULT cuda generator: it is the only one registered
            # with hipGraph capture, so in-graph batch generation (randn
            # included) replays with graph-safe philox state; a custom
            # Generator here made ResNet capture fail in round 1
            # (NOTES.md).  Resume-exactness is unchanged: state_dict
            # snapshots this generator's state like any other.
            idx = device.index if device.index is not None else torch.cuda.current_device()
            self.gen = torch.cuda.default_generators[idx]
        else:
            self.gen = torch.Generator(device=device)
        self.reset()

    def reset(self, step: int = 0) -> None:
        # Seed folds in rank so ranks see distinct data, and step so a
        # resume regenerates the same stream position deterministically.
        self.gen.manual_seed(self.cfg.data.seed * 1000003 + self.rank * 997 + step)

    def state_dict(self) -> dict:
        return {"gen_state": self.gen.get_state()}

    def load_state_dict(self, sd: dict) -> None:
        self.gen.set_state(sd["gen_state"].to(self.gen.get_state().device))

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        while True:
            yield self.next_batch()

    def next_batch(self) -> Tuple[torch.Tensor, torch.Tensor]:
        raise NotImplementedError


class SyntheticImageData(SyntheticBase):
    """ImageNet-shaped batches: (B,3,H,W) images + class labels."""

    def next_batch(self) -> Tuple[torch.Tensor, torch.Tensor]:
        d = self.cfg.data
        x = torch.randn(
            d.batch_size, d.channels, d.image_size, d.image_size,
            generator=self.gen, device=self.device, dtype=torch.float32,
        ).to(self.dtype)
        if self.device.type == "cuda":
            x = x.to(memory_format=torch.channels_last)
        y = torch.randint(
            0, d.num_classes, (d.batch_size,), generator=self.gen, device=self.device
        )
        return x, y


class SyntheticTokenData(SyntheticBase):
    """Token streams for LM configs: (B,S) int64 inputs, (B,S) targets.

    data.active_vocab > 0 restricts sampling to the first N token ids so
    a short training run has learnable signal (loss drops from ~ln(V)
    toward ~ln(N)); 0 = uniform over the full vocab (pure throughput).
    """

    def next_batch(self) -> Tuple[torch.Tensor, torch.Tensor]:
        d = self.cfg.data
        vocab = self.cfg.model.vocab_size
        hi = d.active_vocab if d.active_vocab else vocab
        tokens = torch.randint(
            0, min(hi, vocab), (d.batch_size, d.seq_len + 1), generator=self.gen,
            device=self.device
        )
        return tokens[:, :-1].contiguous(), tokens[:, 1:].contiguous()


class SyntheticVectorData(SyntheticBase):
    """Flat feature vectors for the MLP smoke config (C10).

    Targets follow a fixed random linear map of the inputs so that a
    training run has signal and the loss measurably decreases.
    """

    def __init__(self, cfg: TrainConfig, rank: int, device: torch.device, dtype: torch.dtype):
        super().__init__(cfg, rank, device, dtype)
        teacher_gen = torch.Generator(device=device)
        teacher_gen.manual_seed(cfg.data.seed)  # same teacher on all ranks
        self.teacher = torch.randn(
            cfg.data.vector_dim, cfg.model.out_dim, generator=teacher_gen,
            device=device, dtype=torch.float32,
        )

    def next_batch(self) -> Tuple[torch.Tensor, torch.Tensor]:
        d = self.cfg.data
        x = torch.randn(
            d.batch_size, d.vector_dim, generator=self.gen, device=self.device,
            dtype=torch.float32,
        )
        y = x @ self.teacher
        return x.to(self.dtype), y.to(self.dtype)


def build_data(cfg: TrainConfig, rank: int, device: torch.device, dtype: torch.dtype) -> SyntheticBase:
    kind = cfg.data.kind
    if kind == "synthetic_image":
        return SyntheticImageData(cfg, rank, device, dtype)
    if kind == "synthetic_tokens":
        return SyntheticTokenData(cfg, rank, device, dtype)
    if kind == "synthetic_vector":
        return SyntheticVectorData(cfg, rank, device, dtype)
    if kind == "token_file":
        return TokenFileData(cfg, rank, device, dtype)
    raise ValueError(f"unknown data kind {kind!r}")


class TokenFileData:
    """Real-data LM path: a flat binary corpus of token ids, memory-
    mapped and sampled as random (seq_len+1)-windows per rank.

    File format is tokenizer-agnostic (nanoGPT-style): contiguous
    uint16 ids when model.vocab_size <= 65536, else uint32 — anything
    the wheelhouse `tokenizers`/`sentencepiece` emit can be dumped with
    ndarray.tofile.  Window starts come from a CPU torch.Generator
    seeded with the same (seed, rank, step) fold as the synthetic
    streams, and its state rides the checkpoint's existing
    `rng.data.{rank}` slot (scaffold/ckpt/sharded.py), so resume is
    bit-exact and each rank sees a distinct stream.
    """

    def __init__(self, cfg: TrainConfig, rank: int, device: torch.device,
                 dtype: torch.dtype):
        import numpy as np

        if not cfg.data.path:
            raise ValueError("data.kind=token_file requires data.path")
        self.cfg = cfg
        self.rank = rank
        self.device = device
        nd = np.uint16 if cfg.model.vocab_size <= 65536 else np.uint32
        self.arr = np.memmap(cfg.data.path, dtype=nd, mode="r")
        need = cfg.data.seq_len + 1
        if len(self.arr) < need + 1:
            raise ValueError(
                f"token file {cfg.data.path} has {len(self.arr)} tokens; "
                f"need > {need}")
        self.gen = torch.Generator()
        self.reset()

    def reset(self, step: int = 0) -> None:
        self.gen.manual_seed(
            self.cfg.data.seed * 1000003 + self.rank * 997 + step)

    def state_dict(self) -> dict:
        return {"gen_state": self.gen.get_state()}

    def load_state_dict(self, sd: dict) -> None:
        self.gen.set_state(sd["gen_state"].cpu())

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        while True:
            yield self.next_batch()

    def next_batch(self) -> Tuple[torch.Tensor, torch.Tensor]:
        import numpy as np

        d = self.cfg.data
        S = d.seq_len
        n = len(self.arr) - S - 1
        starts = torch.randint(0, n, (d.batch_size,), generator=self.gen)
        win = np.stack([np.asarray(self.arr[s:s + S + 1], dtype=np.int64)
                        for s in starts.tolist()])
        tokens = torch.from_numpy(win)
        if self.device.type == "cuda":
            tokens = tokens.pin_memory().to(self.device, non_blocking=True)
        return tokens[:, :-1].contiguous(), tokens[:, 1:].contiguous()
