"""Device-resident replay buffers.

The reference keeps buffers on host numpy with full device<->host round trips
every step (``/root/reference/gcbfplus/trainer/buffer.py:29-93``);
here everything stays in HBM (288 GB per GPU) and sampling is an index gather.

Capacity semantics faithfully mirror the reference (quirks included):
  - rollout buffers cap the number of rollout ROWS at ``size``;
  - the flat unsafe-sample buffer caps SAMPLES at ``size`` (i.e.
    buffer_size // 2 ~ 256 most-recent unsafe timesteps), and sampling with
    replacement heavily oversamples them — a load-bearing training behavior.
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

from .data import FlatBatch, Rollout


def _tree_cat2(a, b):
    if isinstance(a, torch.Tensor):
        return torch.cat([a, b], dim=0)
    return type(a)(*[_tree_cat2(x, y) for x, y in zip(a, b)])


def _tree_tail(a, k: int):
    if isinstance(a, torch.Tensor):
        return a[-k:]
    return type(a)(*[_tree_tail(x, k) for x in a])


def _tree_index(a, idx):
    if isinstance(a, torch.Tensor):
        return a[idx]
    return type(a)(*[_tree_index(x, idx) for x in a])


class MaskedRolloutBuffer:
    """Stores whole rollouts (b, T, ...) + safe/unsafe masks (b, T, N)."""

    def __init__(self, size: int):
        self._size = size
        self._data: Optional[Tuple[Rollout, torch.Tensor, torch.Tensor]] = None

    @property
    def length(self) -> int:
        return 0 if self._data is None else self._data[0].length

    @property
    def n_data(self) -> int:
        return 0 if self._data is None else self._data[0].n_data

    def append(self, rollout: Rollout, safe: torch.Tensor, unsafe: torch.Tensor):
        if self._data is None:
            self._data = (rollout, safe, unsafe)
        else:
            r0, s0, u0 = self._data
            self._data = (_tree_cat2(r0, rollout), torch.cat([s0, safe]), torch.cat([u0, unsafe]))
        if self._data[0].length > self._size:
            r, s, u = self._data
            self._data = (_tree_tail(r, self._size), s[-self._size:], u[-self._size:])

    def sample(self, k: int, rng: np.random.Generator):
        """k rollout rows, with replacement (reference buffer.py:86-89)."""
        idx = torch.from_numpy(rng.integers(0, self.length, size=k)).to(
            self._data[1].device
        )
        r, s, u = self._data
        return _tree_index(r, idx), s[idx], u[idx]


class FlatSampleBuffer:
    """Stores flat per-timestep samples (M, ...) — the unsafe-sample buffer."""

    def __init__(self, size: int):
        self._size = size
        self._data: Optional[FlatBatch] = None

    @property
    def length(self) -> int:
        return 0 if self._data is None else self._data.n

    def append(self, batch: FlatBatch):
        if batch.n == 0:
            return
        self._data = batch if self._data is None else FlatBatch.cat([self._data, batch])
        if self._data.n > self._size:
            self._data = self._data[-self._size:]

    def sample(self, k: int, rng: np.random.Generator) -> FlatBatch:
        if self.length == 0:
            raise ValueError("empty buffer")
        idx = torch.from_numpy(rng.integers(0, self.length, size=k)).to(self._data.states.device)
        return self._data[idx]
