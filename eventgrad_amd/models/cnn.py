"""Small CNNs from the reference.

CNN-2 [dmnist/event/event.cpp:51-83] — the MNIST EventGraD model (27,480
params / 8 tensors): conv(1->10,k3) -> pool2 -> relu; conv(10->20,k3) ->
Dropout2d -> pool2 -> relu; fc 500->50 relu; dropout(.5); fc 50->10;
log_softmax.

CNN-1 [dmnist/event/event.cpp:15-48, commented out in the reference but the
EventGraD-paper model]: conv(1->10,k5) -> pool2 -> relu; conv(10->20,k5) ->
Dropout2d -> pool2 -> relu; fc 320->100 relu; dropout; fc 100->10.

LeNet5 [dcifar10/common/nnet.hpp:3-33] — CIFAR alternative model (included
but never instantiated by the reference mains; kept for parity).
"""

from __future__ import annotations

import torch
from torch import nn

from ..ops import functional as O
from .layers import Conv2d, Linear


class _MnistCNN(nn.Module):
    uses_dropout = True  # per-step host RNG -> not hipGraph-capturable

    def __init__(self, k: int, fc_in: int, fc_mid: int):
        super().__init__()
        self.conv1 = Conv2d(1, 10, k)
        self.conv2 = Conv2d(10, 20, k)
        self.fc1 = Linear(fc_in, fc_mid)
        self.fc2 = Linear(fc_mid, 10)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = O.to_compute(x)
        x = O.relu(O.max_pool2x2(self.conv1(x)))
        x = self.conv2(x)
        x = O.dropout2d(x, 0.5, self.training)
        x = O.relu(O.max_pool2x2(x))
        x = O.flatten_features(x)
        x = O.relu(self.fc1(x))
        x = O.dropout(x, 0.5, self.training)
        x = self.fc2(x)
        return O.log_softmax(x) if not self.training else x
        # NOTE: in training the caller uses O.nll_of_logits, which applies
        # log_softmax internally; the reference applies it twice
        # (model + loss), which is a mathematical no-op (idempotent).


class CNN2(_MnistCNN):
    def __init__(self):
        super().__init__(k=3, fc_in=500, fc_mid=50)


class CNN1(_MnistCNN):
    def __init__(self):
        super().__init__(k=5, fc_in=320, fc_mid=100)


class LeNet5(nn.Module):
    uses_dropout = True

    def __init__(self):
        super().__init__()
        self.conv1 = Conv2d(3, 6, 5)
        self.conv2 = Conv2d(6, 16, 5)
        self.fc1 = Linear(16 * 5 * 5, 120)
        self.fc2 = Linear(120, 84)
        self.fc3 = Linear(84, 10)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = O.to_compute(x)
        x = O.max_pool2x2(O.relu(self.conv1(x)))
        x = self.conv2(x)
        x = O.dropout2d(x, 0.5, self.training)
        x = O.max_pool2x2(O.relu(x))
        x = O.flatten_features(x)
        x = O.relu(self.fc1(x))
        x = O.relu(self.fc2(x))
        x = self.fc3(x)
        return O.log_softmax(x) if not self.training else x
