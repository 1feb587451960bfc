"""Flat parameter space + fused-SGD-semantics tests (CPU oracle)."""

import torch

from eventgrad_amd.config import RunConfig
from eventgrad_amd.models import build_model
from eventgrad_amd.parallel.engine import _CpuK
from eventgrad_amd.parallel.flat import FlatParamSpace


def _space(model="cnn2"):
    torch.manual_seed(0)
    m = build_model(model)
    return m, FlatParamSpace(m, torch.device("cpu"))


def test_params_are_views():
    m, space = _space()
    p0 = next(m.parameters())
    space.param.add_(1.0)
    assert torch.allclose(p0.data, space.seg(space.param, 0).view(p0.shape))


def test_backward_fills_flat_grad():
    m, space = _space()
    x = torch.randn(4, 1, 28, 28)
    y = m(x).square().mean()
    y.backward()
    assert space.grad.abs().sum() > 0
    g0 = next(m.parameters()).grad
    assert g0.data_ptr() == space.seg(space.grad, 0).data_ptr()


def test_sgd_matches_torch_optim():
    """Fused flat SGD (momentum 0.9) == torch.optim.SGD step-for-step."""
    torch.manual_seed(0)
    m1 = build_model("mlp")
    torch.manual_seed(0)
    m2 = build_model("mlp")
    space = FlatParamSpace(m1, torch.device("cpu"))
    opt = torch.optim.SGD(m2.parameters(), lr=0.1, momentum=0.9)
    for it in range(5):
        torch.manual_seed(100 + it)
        x = torch.randn(8, 1, 28, 28)
        space.zero_grad()
        m1(x).square().mean().backward()
        opt.zero_grad()
        m2(x).square().mean().backward()
        _CpuK.sgd_step(space, lr=0.1, momentum=0.9, wd=0.0)
        opt.step()
        for (n1, p1), (n2, p2) in zip(m1.named_parameters(),
                                      m2.named_parameters()):
            assert torch.allclose(p1, p2, atol=1e-7), (it, n1)


def test_sqnorms_match_torch_norm():
    m, space = _space("mlp")
    sq = _CpuK.sqnorms(space, space.param)
    for i, (_, p) in enumerate(m.named_parameters()):
        assert torch.allclose(sq[i].sqrt(), p.norm(), atol=1e-5)


def test_pack_unpack_roundtrip():
    m, space = _space("mlp")
    fired = [0, 2]
    payload = _CpuK.pack(space, space.param, fired)
    assert payload.numel() == sum(space.numels[i] for i in fired)
    inbox = space.new_like()
    _CpuK.unpack(space, payload, fired, inbox)
    for i in range(space.sz):
        if i in fired:
            assert torch.equal(space.seg(inbox, i), space.seg(space.param, i))
        else:
            assert space.seg(inbox, i).abs().sum() == 0


def test_avg3_semantics():
    m, space = _space("mlp")
    left = torch.ones_like(space.param) * 2
    right = torch.ones_like(space.param) * 4
    before = space.param.clone()
    _CpuK.avg3(space.param, left, right)
    assert torch.allclose(space.param, (before + 6) / 3)


def test_sgd_weight_decay_matches_torch():
    torch.manual_seed(0)
    m1 = build_model("mlp")
    torch.manual_seed(0)
    m2 = build_model("mlp")
    space = FlatParamSpace(m1, torch.device("cpu"))
    opt = torch.optim.SGD(m2.parameters(), lr=0.05, momentum=0.9,
                          weight_decay=1e-4)
    for it in range(4):
        torch.manual_seed(300 + it)
        x = torch.randn(8, 1, 28, 28)
        space.zero_grad()
        m1(x).square().mean().backward()
        opt.zero_grad()
        m2(x).square().mean().backward()
        _CpuK.sgd_step(space, lr=0.05, momentum=0.9, wd=1e-4)
        opt.step()
        for p1, p2 in zip(m1.parameters(), m2.parameters()):
            assert torch.allclose(p1, p2, atol=1e-7)
