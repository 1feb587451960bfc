"""MNIST MLP used by the cent/decent trainers [dmnist/cent/cent.cpp:16-35]."""

from __future__ import annotations

import torch
from torch import nn

from ..ops import functional as O
from .layers import Linear


class MLP(nn.Module):
    """784 -> 128 -> 10 with ReLU after BOTH layers (as the reference does:
    cent.cpp:27-29 applies relu to fc2's output as well)."""

    def __init__(self):
        super().__init__()
        self.fc1 = Linear(784, 128)
        self.fc2 = Linear(128, 10)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x.reshape(x.shape[0], 784)
        if O.use_native(x):
            x = x.to(torch.bfloat16)
        x = O.relu(self.fc1(x))
        x = O.relu(self.fc2(x))
        # eval-mode log-prob head: the reference's test loop applies
        # nll_loss(log_softmax(prediction)) itself (cent.cpp:192); our
        # evaluate() expects log-probs from eval-mode forward, like
        # cnn/resnet.
        return x if self.training else O.log_softmax(x)
