"""Model-parity tests vs the reference architectures."""

import pytest
import torch

from eventgrad_amd.models import build_model


@pytest.mark.parametrize("name,n_tensors,n_params", [
    ("mlp", 4, 101770),        # cent.cpp MLP 784-128-10
    ("cnn2", 8, 27480),        # event.cpp CNN-2
    ("resnet18q", 86, 17444682),  # quirk ResNet
    ("resnet18", 62, 11173962),   # standard ResNet-18 (CIFAR stem)
    ("resnet20", 65, 272474),     # classic CIFAR ResNet-20 (He et al.)
])
def test_param_counts(name, n_tensors, n_params):
    m = build_model(name)
    ps = list(m.named_parameters())
    assert len(ps) == n_tensors
    assert sum(p.numel() for _, p in ps) == n_params


@pytest.mark.parametrize("name,shape", [
    ("mlp", (2, 1, 28, 28)),
    ("cnn1", (2, 1, 28, 28)),
    ("cnn2", (2, 1, 28, 28)),
    ("lenet5", (2, 3, 32, 32)),
    ("resnet18q", (2, 3, 32, 32)),
    ("resnet50q", (2, 3, 32, 32)),
    ("resnet20", (2, 3, 32, 32)),
])
def test_forward_shapes(name, shape):
    torch.manual_seed(0)
    m = build_model(name)
    y = m(torch.randn(shape))
    assert y.shape == (shape[0], 10)
    assert torch.isfinite(y).all()


def test_backward_produces_grads():
    torch.manual_seed(0)
    m = build_model("resnet18q")
    y = m(torch.randn(2, 3, 32, 32))
    loss = y.square().mean()
    loss.backward()
    for n, p in m.named_parameters():
        assert p.grad is not None, n
        assert torch.isfinite(p.grad).all(), n


def test_bn_buffers_not_parameters():
    """Running stats must not be communicated (ref: named_parameters only)."""
    m = build_model("resnet18q")
    names = [n for n, _ in m.named_parameters()]
    assert not any("running" in n for n in names)
    bufs = [n for n, _ in m.named_buffers()]
    assert any("running_mean" in n for n in bufs)


def test_quirk_vs_standard_block_counts():
    q = build_model("resnet18q")
    s = build_model("resnet18")
    assert len(q.layer1) == 3 and len(s.layer1) == 2
