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
    out = torch.empty((n, x.shape[1], crop, crop), dtype=x.dtype)
    for i in range(n):
        r, c = int(offs[i, 0]), int(offs[i, 1])
        out[i] = xp[i, :, r:r + crop, c:c + crop]
    return out
