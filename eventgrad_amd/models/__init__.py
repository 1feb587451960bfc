"""Model zoo covering every architecture in the reference.

  - mlp       MNIST MLP 784-128-10            [dmnist/cent/cent.cpp:16-35]
  - cnn1      EventGraD-paper CNN-1 (k5)      [dmnist/event/event.cpp:15-48]
  - cnn2      EventGraD-paper CNN-2 (k3)      [dmnist/event/event.cpp:51-83]
  - lenet5    small CIFAR CNN                 [dcifar10/common/nnet.hpp:3-33]
  - resnetNq  quirk ResNets (reference make_layer builds layers[i]+1 blocks
              per stage -> "ResNet-18" = 17,444,682 params / 86 tensors)
                                              [dcifar10/common/resnet.hpp:160-181]
  - resnetN   standard ResNets (layers[i] blocks per stage)
"""

from .layers import Conv2d, BatchNorm2d, Linear  # noqa: F401
from .mlp import MLP  # noqa: F401
from .cnn import CNN1, CNN2, LeNet5  # noqa: F401
from .resnet import ResNet, BasicBlock, BottleNeck  # noqa: F401


def build_model(name: str, num_classes: int = 10):
    from .resnet import resnet_factory

    name = name.lower()
    if name == "mlp":
        return MLP()
    if name == "cnn1":
        return CNN1()
    if name == "cnn2":
        return CNN2()
    if name == "lenet5":
        return LeNet5()
    if name.startswith("resnet"):
        return resnet_factory(name, num_classes)
    raise ValueError(f"unknown model {name!r}")


MODEL_NAMES = ("mlp", "cnn1", "cnn2", "lenet5", "resnet18q", "resnet18",
               "resnet34q", "resnet34", "resnet50q", "resnet50",
               "resnet101", "resnet152", "resnet20", "resnet32",
               "resnet44", "resnet56")
