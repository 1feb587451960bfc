"""Flat fused parameter/grad/momentum space.

The reference packs each tensor separately per message with an element-wise
copy loop (decent.cpp:183-189 — a per-element ``.item<float>()`` hot spot)
and lays the RMA window out as [left-half | right-half] of tightly packed
per-tensor segments (event.cpp:169-179). The MI355X design instead keeps
ONE persistent fp32 flat buffer per role (params / grads / momentum /
left-inbox / right-inbox) with every ``named_parameters()`` tensor re-pointed
to a view of the param buffer, so:

  * the fused HIP SGD-step kernel updates all tensors in one launch and
    emits every per-tensor L2 norm on the way (BASELINE north star);
  * (p+l+r)/3 averaging is one elementwise kernel over the whole space;
  * gossip payloads are packed/unpacked with one gather/scatter kernel.

Segment starts are padded to 64 floats (256 B) so every segment is
vector-load aligned; the pad gaps stay zero and are ignored by the wire
format (payloads are tight-packed).
"""

from __future__ import annotations

from typing import List, Tuple

import torch
from torch import nn

ALIGN = 64  # floats


class FlatParamSpace:
    def __init__(self, model: nn.Module, device: torch.device):
        self.device = device
        self.names: List[str] = []
        shapes = []
        numels = []
        params = []
        for name, p in model.named_parameters():
            self.names.append(name)
            shapes.append(tuple(p.shape))
            numels.append(p.numel())
            params.append(p)
        self.shapes: List[Tuple[int, ...]] = shapes
        self.numels: List[int] = numels
        self.sz = len(params)

        starts = []
        off = 0
        for n in numels:
            starts.append(off)
            off += (n + ALIGN - 1) // ALIGN * ALIGN
        self.starts: List[int] = starts
        self.total: int = off
        self.total_tight: int = sum(numels)

        self.param = torch.zeros(self.total, dtype=torch.float32, device=device)
        self.grad = torch.zeros_like(self.param)
        self.momentum = torch.zeros_like(self.param)

        # device-side segment tables for the HIP kernels
        self.starts_t = torch.tensor(starts, dtype=torch.int64, device=device)
        self.numels_t = torch.tensor(numels, dtype=torch.int64, device=device)

        # re-point model parameters and grads into the flat space
        for p, s, n, shape in zip(params, starts, numels, shapes):
            view = self.param[s:s + n].view(shape)
            with torch.no_grad():
                view.copy_(p.data)
            p.data = view
            p.grad = self.grad[s:s + n].view(shape)

    # -- views ------------------------------------------------------------
    def seg(self, buf: torch.Tensor, i: int) -> torch.Tensor:
        s, n = self.starts[i], self.numels[i]
        return buf[s:s + n]

    def new_like(self) -> torch.Tensor:
        return torch.zeros_like(self.param)

    def clone_params(self) -> torch.Tensor:
        return self.param.clone()

    # -- math (CPU reference path; GPU uses _core kernels via engine) ------
    def sqnorms_cpu(self, buf: torch.Tensor) -> torch.Tensor:
        return torch.stack([self.seg(buf, i).square().sum()
                            for i in range(self.sz)])

    def zero_grad(self) -> None:
        self.grad.zero_()

    def load_flat(self, flat: torch.Tensor) -> None:
        with torch.no_grad():
            self.param.copy_(flat)
