"""Batched augmentations: pad(4) -> random horizontal flip -> random crop(32).

Reference pipeline (dcifar10/event/event.cpp:94-98 using
common/transform.hpp): ConstantPad(4), RandomHorizontalFlip(p=.5),
RandomCrop({32,32}), randomness drawn from the torch RNG (so governed by the
manual seed, transform.hpp:19-101). Implemented batched on NCHW fp32
tensors with an explicit generator for determinism.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F


def augment_batch(x: torch.Tensor, pad: int = 4, crop: int = 32,
                  flip_p: float = 0.5,
                  generator: torch.Generator | None = None) -> torch.Tensor:
    n = x.shape[0]
    xp = F.pad(x, (pad, pad, pad, pad))
    flip = torch.rand(n, generator=generator) < flip_p
    if flip.any():
        xp[flip] = torch.flip(xp[flip], dims=[3])
    max_off = xp.shape[2] - crop
    offs = torch.randint(0, max_off + 1, (n, 2), generator=generator)
    # one batched gather instead of a per-sample python crop loop
    # (the loop throttled end-to-end trainer throughput)
    rows = offs[:, 0:1] + torch.arange(crop)          # [n, crop]
    cols = offs[:, 1:2] + torch.arange(crop)          # [n, crop]
    b = torch.arange(n)[:, None, None]
    return xp[b, :, rows[:, :, None], cols[:, None, :]].permute(0, 3, 1, 2) \
        .contiguous()
