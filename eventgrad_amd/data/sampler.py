"""Distributed samplers: 1 rank = 1 data shard.

Mirrors the LibTorch samplers the reference uses:
  - DistributedSequentialSampler (event.cpp:139-141): contiguous shard,
    fixed order;
  - DistributedRandomSampler (cent.cpp:59-60, spevent.cpp:105-107): epoch-wise
    global shuffle, then shard.
"""

from __future__ import annotations

import numpy as np


class DistributedSequentialSampler:
    def __init__(self, n: int, world: int, rank: int):
        self.n, self.world, self.rank = n, world, rank
        self.per_rank = n // world

    def epoch_indices(self, epoch: int) -> np.ndarray:
        s = self.rank * self.per_rank
        return np.arange(s, s + self.per_rank)


class DistributedRandomSampler:
    def __init__(self, n: int, world: int, rank: int, seed: int = 0):
        self.n, self.world, self.rank, self.seed = n, world, rank, seed
        self.per_rank = n // world

    def epoch_indices(self, epoch: int) -> np.ndarray:
        rng = np.random.default_rng(self.seed * 100003 + epoch)
        perm = rng.permutation(self.n)
        s = self.rank * self.per_rank
        return perm[s:s + self.per_rank]
