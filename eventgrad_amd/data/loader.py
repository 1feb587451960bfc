"""Minimal shard-aware batch loader (replaces torch::data::make_data_loader).

batch_size == 0 means "whole shard per batch" — the reference's full-batch
configs (cent.cpp:62-65, decent.cpp:81-88).
"""

from __future__ import annotations

import numpy as np
import torch

from .transforms import augment_batch


class ShardLoader:
    def __init__(self, dataset, sampler, batch_size: int,
                 augment: bool = False, seed: int = 0,
                 drop_last: bool = True):
        self.dataset = dataset
        self.sampler = sampler
        self.batch_size = batch_size
        self.augment = augment
        self.seed = seed
        self.drop_last = drop_last

    def num_batches(self) -> int:
        per = self.sampler.per_rank
        bs = self.batch_size or per
        return per // bs if self.drop_last else (per + bs - 1) // bs

    def epoch(self, epoch: int):
        idx = self.sampler.epoch_indices(epoch)
        bs = self.batch_size or len(idx)
        g = None
        if self.augment:
            g = torch.Generator().manual_seed(self.seed * 7919 + epoch)
        nb = len(idx) // bs if self.drop_last else (len(idx) + bs - 1) // bs
        for b in range(nb):
            bi = idx[b * bs:(b + 1) * bs]
            x, y = self.dataset.batch(np.asarray(bi))
            if self.augment:
                x = augment_batch(x, generator=g)
            yield x, y
