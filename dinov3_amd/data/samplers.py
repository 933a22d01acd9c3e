"""Samplers: EpochSampler (+ the Infinite/ShardedInfinite samplers the
reference left as comments, implemented for real here).

Parity: dinov3_jax/data/samplers.py:30-67 — tile dataset indices to >= size,
seeded permutation per epoch, strided slice [rank::world].
"""

from __future__ import annotations

import itertools
from typing import Iterator, Optional

import numpy as np
import torch.utils.data

from .. import parallel


class EpochSampler(torch.utils.data.Sampler):
    """Per-epoch seeded permutation, rank-strided. `advance` skips that many
    already-consumed samples on resume: whole epochs raise the effective epoch
    index, the remainder is skipped inside the first epoch iterated (the
    reference forces EPOCH sampling but raises NotImplementedError for
    advance>0, dinov3_jax/data/loaders.py:133-134 — resume needs it real)."""

    def __init__(self, *, size: int, sample_count: int, shuffle: bool = False, seed: int = 0,
                 start: Optional[int] = None, step: Optional[int] = None, advance: int = 0):
        self._size = size
        self._sample_count = sample_count
        self._shuffle = shuffle
        self._seed = seed
        self._start = parallel.get_rank() if start is None else start
        self._step = parallel.get_world_size() if step is None else step
        self._epoch = 0
        per_epoch = max(len(self), 1)
        self._epoch_offset = advance // per_epoch
        self._skip_first = advance % per_epoch

    def __len__(self) -> int:
        return (self._size - self._start + self._step - 1) // self._step

    def _iterable(self):
        count = (self._size + self._sample_count - 1) // self._sample_count
        tiled = np.tile(np.arange(self._sample_count), count)[: self._size]
        if self._shuffle:
            rng = np.random.default_rng(self._seed + self._epoch + self._epoch_offset)
            tiled = rng.permutation(tiled)
        return tiled[self._start:: self._step]

    def __iter__(self) -> Iterator[int]:
        it = map(int, self._iterable())
        if self._skip_first > 0:
            it = itertools.islice(it, self._skip_first, None)
            self._skip_first = 0
        yield from it

    def set_epoch(self, epoch: int) -> None:
        self._epoch = epoch


class InfiniteSampler(torch.utils.data.Sampler):
    def __init__(self, *, sample_count: int, shuffle: bool = False, seed: int = 0,
                 start: Optional[int] = None, step: Optional[int] = None, advance: int = 0):
        self._sample_count = sample_count
        self._shuffle = shuffle
        self._seed = seed
        self._start = parallel.get_rank() if start is None else start
        self._step = parallel.get_world_size() if step is None else step
        self._advance = advance

    def _gen(self):
        if self._shuffle:
            rng = np.random.default_rng(self._seed)
            while True:
                yield from map(int, rng.permutation(self._sample_count))
        else:
            while True:
                yield from range(self._sample_count)

    def __iter__(self) -> Iterator[int]:
        it = itertools.islice(self._gen(), self._start + self._advance * self._step, None, self._step)
        yield from it


class ShardedInfiniteSampler(torch.utils.data.Sampler):
    """Infinite sampler that re-shuffles with a new seed every pass and keeps
    each rank's shard disjoint within a pass."""

    def __init__(self, *, sample_count: int, shuffle: bool = True, seed: int = 0,
                 start: Optional[int] = None, step: Optional[int] = None, advance: int = 0):
        self._sample_count = sample_count
        self._shuffle = shuffle
        self._seed = seed
        self._start = parallel.get_rank() if start is None else start
        self._step = parallel.get_world_size() if step is None else step
        self._advance = advance

    def __iter__(self) -> Iterator[int]:
        epoch = 0
        emitted = 0
        while True:
            if self._shuffle:
                rng = np.random.default_rng(self._seed + epoch)
                perm = rng.permutation(self._sample_count)
            else:
                perm = np.arange(self._sample_count)
            shard = perm[self._start:: self._step]
            for idx in shard:
                if emitted >= self._advance:
                    yield int(idx)
                emitted += 1
            epoch += 1
