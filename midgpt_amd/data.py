"""Data loading: uint16 token bins + random-window batching.

Parity with the reference's get_batch / per-process split
(src/train.py:56-66, 122-124): uniform-random window starts over a
contiguous per-process slab, x = data[i:i+T], y = data[i+1:i+T+1],
batched to (G, B_local, T). Unseeded by default (reference parity);
pass a seeded numpy Generator for reproducible streams.

A synthetic mode serves the benchmark contract (no network for datasets):
random tokens of the right shape/vocab.
"""
from __future__ import annotations

import os

import numpy as np
import torch


def load_bin(path: str, copy_to_ram: bool = True) -> np.ndarray:
    """Load a uint16 token bin (memmap; optionally copied into RAM as the
    reference does, src/train.py:132-133)."""
    arr = np.memmap(path, dtype=np.uint16, mode="r")
    return np.array(arr) if copy_to_ram else arr


def split_by_process(data: np.ndarray, rank: int, world: int) -> np.ndarray:
    """Contiguous per-process split (reference src/train.py:122-124)."""
    n = len(data) // world
    return data[rank * n:(rank + 1) * n]


def get_batch(data: np.ndarray, block_size: int, batch_size: int,
              g_accum_iters: int, rng: np.random.Generator | None = None
              ) -> tuple[torch.Tensor, torch.Tensor]:
    """Returns x, y of shape (G, B, T) int64."""
    nwin = batch_size * g_accum_iters
    hi = len(data) - block_size - 1
    ix = (rng.integers(0, hi, size=nwin) if rng is not None
          else np.random.randint(0, hi, size=nwin))
    x = np.stack([np.asarray(data[i:i + block_size], dtype=np.int64) for i in ix])
    y = np.stack([np.asarray(data[i + 1:i + 1 + block_size], dtype=np.int64) for i in ix])
    shp = (g_accum_iters, batch_size, block_size)
    return torch.from_numpy(x).reshape(shp), torch.from_numpy(y).reshape(shp)


def synthetic_batch(vocab_size: int, block_size: int, batch_size: int,
                    g_accum_iters: int, device=None,
                    generator: torch.Generator | None = None
                    ) -> tuple[torch.Tensor, torch.Tensor]:
    """Random tokens of the training shape (benchmark / smoke use)."""
    tok = torch.randint(0, vocab_size,
                        (g_accum_iters, batch_size, block_size + 1),
                        dtype=torch.int64, generator=generator)
    x, y = tok[..., :-1], tok[..., 1:]
    if device is not None:
        x, y = x.to(device), y.to(device)
    return x, y


class BatchLoader:
    """Unified loader: bins if data_dir has train.bin/val.bin, else synthetic."""

    def __init__(self, data_dir: str, vocab_size: int, block_size: int,
                 rank: int = 0, world: int = 1, synthetic: bool = False,
                 seed: int | None = None):
        self.block_size = block_size
        self.vocab_size = vocab_size
        self.synthetic = synthetic or not os.path.exists(
            os.path.join(data_dir, "train.bin"))
        self.rng = np.random.default_rng(seed + rank) if seed is not None else None
        self.torch_gen = None
        if seed is not None:
            self.torch_gen = torch.Generator().manual_seed(seed + rank)
        if not self.synthetic:
            self.train = split_by_process(
                load_bin(os.path.join(data_dir, "train.bin")), rank, world)
            self.val = split_by_process(
                load_bin(os.path.join(data_dir, "val.bin")), rank, world)

    def batch(self, split: str, batch_size: int, g_accum_iters: int = 1):
        if self.synthetic:
            return synthetic_batch(self.vocab_size, self.block_size, batch_size,
                                   g_accum_iters, generator=self.torch_gen)
        data = self.train if split == "train" else self.val
        return get_batch(data, self.block_size, batch_size, g_accum_iters, self.rng)
