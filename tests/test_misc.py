"""Config/CLI/data/metrics/trace unit tests."""

import os

import numpy as np
import torch

from eventgrad_amd.config import RunConfig, preset
from eventgrad_amd.data import (Cifar10Dataset, DistributedRandomSampler,
                                DistributedSequentialSampler, MnistDataset,
                                SyntheticImages)
from eventgrad_amd.data.loader import ShardLoader
from eventgrad_amd.data.transforms import augment_batch
from eventgrad_amd.train.cli import build_parser, config_from_args
from eventgrad_amd.train.metrics import RunMetrics


def test_presets_roundtrip():
    for name in ("dmnist-cent", "dmnist-event", "dcifar10-event",
                 "dcifar10-spevent"):
        cfg = preset(name)
        s = cfg.to_json()
        cfg2 = RunConfig.from_json(s)
        assert cfg2 == cfg


def test_cli_maps_reference_args():
    p = build_parser()
    args = p.parse_args(["--preset", "dcifar10-event", "--thres-constant",
                         "0.002", "--epochs", "3", "--topk-percent", "2.5"])
    cfg = config_from_args(args)
    assert cfg.mode == "event" and cfg.model == "resnet18q"
    assert not cfg.trigger.adaptive and cfg.trigger.constant == 0.002
    assert cfg.epochs == 3 and cfg.topk_percent == 2.5
    assert cfg.data.global_batch == 256
    assert cfg.optim.momentum == 0.9


def test_synthetic_deterministic_and_learnable():
    ds = SyntheticImages((3, 32, 32), 100, noise=0.3)
    x1, y1 = ds[7]
    x2, y2 = ds[7]
    assert torch.equal(x1, x2) and y1 == y2
    # same-class samples closer than cross-class on average
    xa, _ = ds[0]
    xb, _ = ds[10]  # same class (10 % 10 == 0)
    xc, _ = ds[1]
    assert (xa - xb).norm() < (xa - xc).norm()


def test_samplers_shard_disjoint():
    seq = [DistributedSequentialSampler(100, 4, r) for r in range(4)]
    allidx = np.concatenate([s.epoch_indices(0) for s in seq])
    assert len(set(allidx.tolist())) == 100
    rnd = [DistributedRandomSampler(100, 4, r, seed=0) for r in range(4)]
    allidx = np.concatenate([s.epoch_indices(3) for s in rnd])
    assert len(set(allidx.tolist())) == 100
    assert not np.array_equal(rnd[0].epoch_indices(0),
                              rnd[0].epoch_indices(1))


def test_loader_batches_and_augment():
    ds = SyntheticImages((3, 32, 32), 64)
    s = DistributedSequentialSampler(64, 2, 0)
    ld = ShardLoader(ds, s, 8, augment=True, seed=0)
    batches = list(ld.epoch(1))
    assert len(batches) == 4
    x, y = batches[0]
    assert x.shape == (8, 3, 32, 32) and y.shape == (8,)


def test_augment_shapes_and_determinism():
    x = torch.randn(4, 3, 32, 32)
    g1 = torch.Generator().manual_seed(5)
    g2 = torch.Generator().manual_seed(5)
    a = augment_batch(x, generator=g1)
    b = augment_batch(x, generator=g2)
    assert a.shape == x.shape and torch.equal(a, b)


def test_messages_saved_math():
    m = RunMetrics(world=4, num_tensors=8, total_passes=100,
                   num_events_total=4 * 2 * 8 * 30)  # only warmup fired
    assert abs(m.messages_saved_pct - 70.0) < 1e-9


def test_mnist_idx_parser(tmp_path):
    import struct
    imgs = np.random.randint(0, 255, (10, 28, 28), dtype=np.uint8)
    lbls = np.random.randint(0, 10, (10,), dtype=np.uint8)
    with open(os.path.join(tmp_path, "train-images-idx3-ubyte"), "wb") as f:
        f.write(struct.pack(">BBBBIII", 0, 0, 8, 3, 10, 28, 28))
        f.write(imgs.tobytes())
    with open(os.path.join(tmp_path, "train-labels-idx1-ubyte"), "wb") as f:
        f.write(struct.pack(">BBBBI", 0, 0, 8, 1, 10))
        f.write(lbls.tobytes())
    ds = MnistDataset(str(tmp_path), train=True)
    assert len(ds) == 10
    x, y = ds[3]
    assert x.shape == (1, 28, 28) and y == int(lbls[3])


def test_cifar_binary_parser(tmp_path):
    raw = np.random.randint(0, 255, (5, 3073), dtype=np.uint8)
    raw[:, 0] = np.arange(5) % 10
    for i in range(1, 6):
        raw.tofile(os.path.join(tmp_path, f"data_batch_{i}.bin"))
    ds = Cifar10Dataset(str(tmp_path), train=True)
    assert len(ds) == 25
    x, y = ds[0]
    assert x.shape == (3, 32, 32) and 0 <= y < 10


def test_compute_dtype_plumbing():
    """RunConfig.compute_dtype is consumed: it toggles the ops dispatch
    (the flag must not be dead)."""
    import pytest
    import torch
    from eventgrad_amd.ops import functional as O

    assert O.get_compute_dtype() == "bf16"
    O.set_compute_dtype("fp32")
    try:
        assert O.get_compute_dtype() == "fp32"
        # CPU tensors never take the native path in either mode
        assert not O.use_native(torch.zeros(2))
        from eventgrad_amd.train.graphstep import can_graph

        class _M:
            uses_dropout = False
        assert not can_graph(_M(), torch.device("cpu"))
        with pytest.raises(ValueError):
            O.set_compute_dtype("fp16")
    finally:
        O.set_compute_dtype("bf16")


def test_cli_compute_dtype_flag():
    from eventgrad_amd.train.cli import build_parser, config_from_args
    args = build_parser().parse_args(["--compute-dtype", "fp32"])
    assert config_from_args(args).compute_dtype == "fp32"


def test_jpeg_dir_cifar10(tmp_path):
    """Reference JPEG-dir contract (dcifar10/common/custom.hpp:26-122):
    train/<class>/NNNN.jpg tree, RGB CHW float 0..255, labels by dir."""
    import numpy as np
    import torch
    from PIL import Image
    from eventgrad_amd.data.datasets import (CIFAR10_CLASSES, JpegDirCifar10,
                                             build_dataset)
    from eventgrad_amd.config import DataConfig

    rng = np.random.default_rng(0)
    for split, n in (("train", 3), ("test", 2)):
        for j, cls in enumerate(CIFAR10_CLASSES):
            d = tmp_path / split / cls
            d.mkdir(parents=True)
            for i in range(n):
                # class-coded color so channel ORDER is checkable: class 0
                # strongly red, class 1 strongly green
                base = np.zeros((32, 32, 3), np.uint8)
                base[..., j % 3] = 200
                base += rng.integers(0, 30, base.shape).astype(np.uint8)
                Image.fromarray(base).save(d / f"{i:04d}.jpg", quality=95)

    ds = JpegDirCifar10(str(tmp_path), train=True)
    assert len(ds) == 30
    x, y = ds[0]
    assert x.shape == (3, 32, 32) and x.dtype == torch.float32
    assert y == 0
    # reference keeps raw 0..255 floats (no normalization)
    assert x.max() > 150
    # class 0 images are red-dominant: channel 0 (R) strongest
    assert x[0].mean() > x[1].mean() and x[0].mean() > x[2].mean()
    xb, yb = ds.batch(np.array([0, 3, 6]))
    assert xb.shape == (3, 3, 32, 32)
    assert yb.tolist() == [0, 1, 2]
    # class 1 (automobile) green-dominant in the batch
    assert xb[1, 1].mean() > xb[1, 0].mean()

    # build_dataset auto-detects the layout
    cfg = DataConfig(dataset="cifar10", data_path=str(tmp_path))
    d2 = build_dataset(cfg, train=False)
    assert isinstance(d2, JpegDirCifar10) and len(d2) == 20


def test_trainer_on_jpeg_dir_dataset(tmp_path):
    """End-to-end: the reference's JPEG-dir CIFAR contract feeds the
    Trainer (serial, 1 epoch) through build_dataset auto-detection."""
    import numpy as np
    from PIL import Image
    from eventgrad_amd.config import DataConfig, OptimConfig, RunConfig
    from eventgrad_amd.data.datasets import CIFAR10_CLASSES
    from eventgrad_amd.train.trainer import Trainer

    rng = np.random.default_rng(1)
    for split, n in (("train", 8), ("test", 4)):
        for j, cls in enumerate(CIFAR10_CLASSES):
            d = tmp_path / split / cls
            d.mkdir(parents=True)
            for i in range(n):
                base = np.zeros((32, 32, 3), np.uint8)
                base[..., j % 3] = 150 + 10 * (j // 3)
                base += rng.integers(0, 20, base.shape).astype(np.uint8)
                Image.fromarray(base).save(d / f"{i:04d}.jpg", quality=92)

    cfg = RunConfig(mode="serial", model="lenet5", epochs=1, device="cpu",
                    data=DataConfig(dataset="cifar10",
                                    data_path=str(tmp_path), batch_size=16),
                    optim=OptimConfig(lr=1e-3), eval_at_end=True)
    tr = Trainer(cfg)
    m = tr.train()
    assert m.total_passes == 5            # 80 train images / 16
    assert np.isfinite(m.final_train_loss)
    assert m.test_accuracy is not None    # eval over the 40 test JPEGs
