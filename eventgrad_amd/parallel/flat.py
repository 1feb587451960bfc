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

        self._build_shadows(model, params)

    # -- persistent bf16 weight shadows (GPU native path) ------------------
    # Every conv/linear weight keeps two bf16 shadow copies in kernel-ready
    # layouts: KRSC (conv fwd operand / linear [N,K]) and CRSK (conv dgrad
    # operand / linear [K,N] transpose). ONE kernel launch regenerates all
    # of them from the fp32 flat params after each mutation (fused SGD
    # step, consensus allreduce, checkpoint load), replacing the per-conv
    # per-step oihw_to_krsc / krsc_to_crsk transform launches inside the
    # captured fwd+bwd graph.
    def _build_shadows(self, model, params) -> None:
        self._shadow_meta = None
        if self.device.type != "cuda":
            return
        from ..models.layers import Conv2d as EgConv2d
        from ..models.layers import Linear as EgLinear

        by_param = {id(p): i for i, p in enumerate(params)}
        entries = []  # (seg_idx, K, C, R, S, module)
        for mod in model.modules():
            if isinstance(mod, EgConv2d):
                i = by_param.get(id(mod.weight))
                if i is None:
                    continue
                K, C, R, S = self.shapes[i]
                entries.append((i, K, C, R, S, mod))
            elif isinstance(mod, EgLinear):
                i = by_param.get(id(mod.weight))
                if i is None:
                    continue
                N, Kin = self.shapes[i]
                entries.append((i, N, Kin, 1, 1, mod))
        if not entries:
            return
        off = 0
        sh_starts = []
        for (i, K, C, R, S, mod) in entries:
            sh_starts.append(off)
            off += (K * C * R * S + ALIGN - 1) // ALIGN * ALIGN
        dev = self.device
        bf16 = dict(dtype=torch.bfloat16, device=dev)
        self.shadow_krsc = torch.zeros(off, **bf16)
        self.shadow_crsk = torch.zeros(off, **bf16)
        i64 = dict(dtype=torch.int64, device=dev)
        i32 = dict(dtype=torch.int32, device=dev)
        self._sh_pstart = torch.tensor([self.starts[e[0]] for e in entries],
                                       **i64)
        self._sh_start = torch.tensor(sh_starts, **i64)
        self._sh_K = torch.tensor([e[1] for e in entries], **i32)
        self._sh_C = torch.tensor([e[2] for e in entries], **i32)
        self._sh_R = torch.tensor([e[3] for e in entries], **i32)
        self._sh_S = torch.tensor([e[4] for e in entries], **i32)
        for (i, K, C, R, S, mod), so in zip(entries, sh_starts):
            n = K * C * R * S
            wk = self.shadow_krsc[so:so + n]
            wt = self.shadow_crsk[so:so + n]
            if isinstance(mod, EgConv2d):
                mod._shadow_wk = wk.view(K, R, S, C)
                mod._shadow_wt = wt.view(C, R, S, K)
            else:  # Linear: [N, Kin] and its transpose
                mod._shadow_wk = wk.view(K, C)
                mod._shadow_wt = wt.view(C, K)
        self._shadow_meta = True
        self.refresh_shadows()

    def refresh_shadows(self) -> None:
        """Regenerate every weight shadow from the fp32 flat params (one
        kernel). Call after any param mutation outside fwd/bwd."""
        if self._shadow_meta is None:
            return
        from ..ops import functional as O
        if O.get_compute_dtype() != "bf16":
            return
        from ..ops.backend import native
        native().refresh_conv_shadows(
            self.param, self._sh_pstart, self._sh_start, self._sh_K,
            self._sh_C, self._sh_R, self._sh_S, self.shadow_krsc,
            self.shadow_crsk)

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
        self.refresh_shadows()
