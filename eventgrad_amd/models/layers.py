"""nn.Module layers backed by eventgrad_amd.ops.

Each layer holds parameters in standard PyTorch layouts/dtypes (fp32, OIHW
conv weights) so ``named_parameters()`` matches the reference's communication
unit (flat fp32 per-tensor messages), and dispatches compute
to the HIP NHWC/bf16 path on GPU or plain torch on CPU.

BatchNorm running stats are buffers, not parameters — like the reference they
are never communicated (SURVEY.md §2.6 note on BatchNorm2d).
"""

from __future__ import annotations

import math

import torch
from torch import nn

from ..ops import functional as O


class Conv2d(nn.Module):
    def __init__(self, in_ch, out_ch, kernel_size, stride=1, padding=0,
                 bias=True):
        super().__init__()
        self.in_ch, self.out_ch = in_ch, out_ch
        self.kernel_size, self.stride, self.padding = kernel_size, stride, padding
        self.weight = nn.Parameter(
            torch.empty(out_ch, in_ch, kernel_size, kernel_size))
        self.bias = nn.Parameter(torch.empty(out_ch)) if bias else None
        self.reset_parameters()

    def reset_parameters(self):
        # LibTorch Conv2d default init: kaiming_uniform(a=sqrt(5)) + uniform bias
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        if self.bias is not None:
            fan_in = self.in_ch * self.kernel_size * self.kernel_size
            bound = 1.0 / math.sqrt(fan_in)
            nn.init.uniform_(self.bias, -bound, bound)

    def forward(self, x, bn_stats: bool = False, dgrad_stats_bn=None):
        # fused-BN-stats epilogue requires out_ch % 64 == 0 (csrc/conv.hip);
        # the paired BatchNorm downgrades stats_ready by the same condition
        return O.conv2d(x, self.weight, self.bias, self.stride, self.padding,
                        bn_stats=bn_stats and O.use_native(x) and self.training
                        and self.out_ch % 64 == 0,
                        wk=getattr(self, "_shadow_wk", None),
                        wt=getattr(self, "_shadow_wt", None),
                        dgrad_stats_bn=dgrad_stats_bn)

    def extra_repr(self):
        return (f"{self.in_ch}, {self.out_ch}, k={self.kernel_size}, "
                f"stride={self.stride}, pad={self.padding}, "
                f"bias={self.bias is not None}")


class BatchNorm2d(nn.Module):
    def __init__(self, num_features, eps=1e-5, momentum=0.1):
        super().__init__()
        self.num_features, self.eps, self.momentum = num_features, eps, momentum
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked",
                             torch.zeros(1, dtype=torch.long))
        self._nbt = 0  # python-side mirror; avoids one GPU kernel per step

    def _save_to_state_dict(self, destination, prefix, keep_vars):
        self.num_batches_tracked.fill_(self._nbt)
        super()._save_to_state_dict(destination, prefix, keep_vars)

    def _load_from_state_dict(self, *args, **kwargs):
        super()._load_from_state_dict(*args, **kwargs)
        self._nbt = int(self.num_batches_tracked.item())

    # checkpoint.py hooks (named_buffers() bypasses the state-dict hooks)
    def sync_buffers_for_save(self):
        self.num_batches_tracked.fill_(self._nbt)

    def sync_buffers_after_load(self):
        self._nbt = int(self.num_batches_tracked.item())

    def forward(self, x, fuse_relu: bool = False, stats_ready: bool = False,
                stats_consumer: bool = False):
        if self.training:
            self._nbt += 1
        return O.batch_norm(x, self.weight, self.bias, self.running_mean,
                            self.running_var, self.training, self.momentum,
                            self.eps, fuse_relu,
                            stats_ready and O.use_native(x) and self.training
                            and self.num_features % 64 == 0,
                            stash_module=self if stats_consumer else None)


class Linear(nn.Module):
    def __init__(self, in_features, out_features, bias=True):
        super().__init__()
        self.in_features, self.out_features = in_features, out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.empty(out_features)) if bias else None
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        if self.bias is not None:
            bound = 1.0 / math.sqrt(self.in_features)
            nn.init.uniform_(self.bias, -bound, bound)

    def forward(self, x):
        return O.linear(x, self.weight, self.bias,
                        wk=getattr(self, "_shadow_wk", None),
                        wt=getattr(self, "_shadow_wt", None))

    def extra_repr(self):
        return f"{self.in_features}, {self.out_features}"
