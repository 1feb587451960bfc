"""Datasets, sharded sampling and augmentation.

Covers the reference's data layer: MNIST via the standard
IDX files (torch::data::datasets::MNIST equivalent, cent.cpp:54-56), CIFAR-10
via the standard binary batches (the reference reads per-class JPEG dirs with
OpenCV, dcifar10/common/custom.hpp:26-122 — this environment has no image
decoder, so the binary distribution format is supported instead), plus
deterministic synthetic datasets (class-prototype + noise) for the
no-network benchmark environment.
"""

from .datasets import SyntheticImages, MnistDataset, Cifar10Dataset, build_dataset  # noqa: F401
from .sampler import DistributedSequentialSampler, DistributedRandomSampler  # noqa: F401
from .loader import ShardLoader  # noqa: F401
from . import transforms  # noqa: F401
