"""GPU end-to-end: model fwd/bwd parity vs CPU fp32, serial training smoke,
and the native-extension-actually-loaded guard."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_native_extension_loaded_from_tree():
    import eventgrad_amd._core as c
    assert "eventgrad_amd" in c.__file__
    # ops on GPU must route through it: a conv on cuda with the module
    # present must not raise
    from eventgrad_amd.ops import functional as O
    x = torch.randn(1, 4, 4, 8, device="cuda").to(torch.bfloat16)
    w = torch.randn(8, 8, 3, 3, device="cuda")
    y = O.conv2d(x, w, None, 1, 1)
    assert y.shape == (1, 4, 4, 8)


@pytest.mark.parametrize("model_name", ["cnn2", "resnet18q"])
def test_model_fwd_bwd_parity(model_name):
    """GPU bf16 model vs CPU fp32 model with identical weights."""
    from eventgrad_amd.models import build_model
    from eventgrad_amd.ops import functional as O

    torch.manual_seed(0)
    m_cpu = build_model(model_name)
    torch.manual_seed(0)
    m_gpu = build_model(model_name).to("cuda")
    # identical weights
    for pc, pg in zip(m_cpu.parameters(), m_gpu.parameters()):
        assert torch.allclose(pc, pg.cpu())

    shape = (8, 1, 28, 28) if model_name == "cnn2" else (8, 3, 32, 32)
    torch.manual_seed(1)
    x = torch.randn(shape)
    y = torch.randint(0, 10, (shape[0],))

    m_cpu.train()
    m_gpu.train()
    # disable dropout randomness differences by eval'ing dropout via p=0:
    # CNN2 uses dropout; parity check therefore runs in eval-forward but
    # with training-mode BN? Simplest: compare in train mode for resnet
    # (no dropout) and eval mode for cnn2.
    if model_name == "cnn2":
        m_cpu.eval()
        m_gpu.eval()
        lc = m_cpu(x)
        lg = m_gpu(x.cuda())
        err = (lg.float().cpu() - lc).norm() / lc.norm()
        assert err < 0.05, err.item()
        return

    logits_c = m_cpu(x)
    loss_c = O.nll_of_logits(logits_c, y)
    loss_c.backward()
    logits_g = m_gpu(x.cuda())
    loss_g = O.nll_of_logits(logits_g, y.cuda())
    loss_g.backward()
    assert abs(loss_g.item() - loss_c.item()) / abs(loss_c.item()) < 0.05

    # Gradient direction must agree to within bf16 compounding. Control
    # experiment: a pure-PyTorch bf16 CPU model vs the fp32 model on this
    # exact input gives mean cosine 0.898 / min 0.786 (26 conv layers), so
    # the HIP kernels are held to the same band, not to fp32.
    cos_all = []
    for (n, pc), pg in zip(m_cpu.named_parameters(), m_gpu.parameters()):
        gc = pc.grad.flatten()
        gg = pg.grad.cpu().flatten()
        cos = torch.dot(gc, gg) / (gc.norm() * gg.norm() + 1e-12)
        cos_all.append(cos.item())
    assert np.mean(cos_all) > 0.85, np.mean(cos_all)
    assert min(cos_all) > 0.70, (min(cos_all),
                                 [n for (n, _), c in
                                  zip(m_cpu.named_parameters(), cos_all)
                                  if c < 0.8])


def test_serial_training_learns_gpu():
    from eventgrad_amd.config import DataConfig, OptimConfig, RunConfig
    from eventgrad_amd.train.trainer import Trainer

    cfg = RunConfig(
        mode="serial", model="cnn2", epochs=3, device="cuda",
        data=DataConfig(dataset="synthetic-mnist", batch_size=64,
                        synthetic_train_samples=512,
                        synthetic_test_samples=128, synthetic_noise=0.3),
        optim=OptimConfig(lr=0.05), eval_at_end=False)
    tr = Trainer(cfg)
    m = tr.train()
    assert np.isfinite(m.final_train_loss)
    assert m.epoch_train_acc[-1] > m.epoch_train_acc[0]


def test_resnet_training_step_gpu():
    from eventgrad_amd.config import DataConfig, OptimConfig, RunConfig
    from eventgrad_amd.train.trainer import Trainer

    cfg = RunConfig(
        mode="serial", model="resnet18q", epochs=1, device="cuda",
        data=DataConfig(dataset="synthetic", batch_size=32,
                        synthetic_train_samples=64, synthetic_test_samples=32),
        optim=OptimConfig(lr=0.01, momentum=0.9), eval_at_end=False)
    tr = Trainer(cfg)
    m = tr.train()
    assert np.isfinite(m.final_train_loss)


def test_hipgraph_step_matches_eager():
    """FwdBwdGraph (capture+replay) produces the same gradients as eager on
    the same batch, to within the run-to-run atomic-reduction noise floor
    (wgrad split-K atomics make even eager-vs-eager nondeterministic)."""
    import torch
    from eventgrad_amd.models import build_model
    from eventgrad_amd.ops import functional as O
    from eventgrad_amd.parallel.flat import FlatParamSpace
    from eventgrad_amd.train.graphstep import FwdBwdGraph

    dev = torch.device("cuda")
    torch.manual_seed(0)
    model = build_model("resnet18q").to(dev)
    model.train()
    space = FlatParamSpace(model, dev)
    torch.manual_seed(7)
    x = torch.randn(8, 3, 32, 32, device=dev)
    y = torch.randint(0, 10, (8,), device=dev)

    def eager():
        space.zero_grad()
        loss = O.nll_of_logits(model(x), y)
        loss.backward()
        return float(loss.detach()), space.grad.clone()

    l1, g1 = eager()
    l2, g2 = eager()
    floor = ((g1 - g2).norm() / g1.norm()).item()  # atomic noise floor

    graph = FwdBwdGraph(model, space, (8, 3, 32, 32), dev)
    _, loss_g = graph.step(x, y)
    torch.cuda.synchronize()
    lg = float(loss_g.detach())
    gg = space.grad.clone()
    err = ((gg - g1).norm() / g1.norm()).item()
    assert abs(lg - l1) / max(abs(l1), 1e-6) < 1e-2, (lg, l1)
    assert err < max(10 * floor, 5e-3), (err, floor)
    # replay a second batch and check it responds to the new input
    x2 = torch.randn_like(x)
    _, loss2 = graph.step(x2, y)
    torch.cuda.synchronize()
    assert abs(float(loss2.detach()) - lg) > 1e-6  # input actually flows


def test_gpu_checkpoint_roundtrip(tmp_path):
    """Checkpoint/resume with the device-resident controller state."""
    import os
    from eventgrad_amd.config import DataConfig, OptimConfig, RunConfig
    from eventgrad_amd.train.trainer import Trainer

    ck = os.path.join(tmp_path, "ck.pt")

    def cfg(epochs, resume):
        c = RunConfig(
            mode="serial", model="cnn2", epochs=epochs, device="cuda",
            data=DataConfig(dataset="synthetic-mnist", batch_size=64,
                            synthetic_train_samples=256,
                            synthetic_test_samples=64),
            optim=OptimConfig(lr=0.05, momentum=0.9), eval_at_end=False,
            checkpoint_path=ck, resume=resume, final_consensus=False)
        return c

    tr1 = Trainer(cfg(1, False))
    tr1.train()
    p_after1 = tr1.space.param.clone()
    tr2 = Trainer(cfg(2, True))
    assert torch.allclose(tr2.space.param, p_after1)
    m = tr2.train()
    assert tr2.pass_num == 8  # 4 passes/epoch x 2 epochs total
    assert np.isfinite(m.final_train_loss)


def test_resnet20_gpu_step():
    """Classic CIFAR ResNet-20 on GPU (16/32-channel convs exercise the
    fused-stats downgrade path: K % 64 != 0 -> separate BN stats)."""
    from eventgrad_amd.models import build_model
    from eventgrad_amd.ops import functional as O
    from eventgrad_amd.parallel.flat import FlatParamSpace
    from eventgrad_amd.ops.backend import native

    dev = torch.device("cuda")
    torch.manual_seed(0)
    m = build_model("resnet20").to(dev)
    m.train()
    space = FlatParamSpace(m, dev)
    x = torch.randn(8, 3, 32, 32, device=dev)
    y = torch.randint(0, 10, (8,), device=dev)
    loss = O.nll_of_logits(m(x), y)
    loss.backward()
    native().sgd_step_norm(space.param, space.grad, space.momentum,
                           space.starts_t, space.numels_t, 0.01, 0.9, 0.0)
    torch.cuda.synchronize()
    assert torch.isfinite(loss).item()
    assert bool(torch.isfinite(space.grad).all().item())


def test_bottleneck_resnet_gpu_step():
    """BottleNeck (1-3-1) block path on GPU: one fwd/bwd/step, finite."""
    from eventgrad_amd.models import build_model
    from eventgrad_amd.ops import functional as O
    from eventgrad_amd.parallel.flat import FlatParamSpace
    from eventgrad_amd.ops.backend import native

    dev = torch.device("cuda")
    torch.manual_seed(0)
    m = build_model("resnet50q").to(dev)
    m.train()
    space = FlatParamSpace(m, dev)
    x = torch.randn(4, 3, 32, 32, device=dev)
    y = torch.randint(0, 10, (4,), device=dev)
    loss = O.nll_of_logits(m(x), y)
    loss.backward()
    norms = native().sgd_step_norm(space.param, space.grad, space.momentum,
                                   space.starts_t, space.numels_t,
                                   0.01, 0.9, 0.0)
    torch.cuda.synchronize()
    assert torch.isfinite(loss).item()
    assert bool(torch.isfinite(norms).all().item())
    assert bool(torch.isfinite(space.grad).all().item())


@pytest.mark.parametrize("name", ["mlp", "cnn1", "cnn2", "lenet5",
                                  "resnet18", "resnet34q", "resnet50",
                                  "resnet32", "resnet101"])
def test_model_zoo_gpu_step(name):
    """Every zoo architecture: one fwd/bwd/step on GPU, finite results."""
    from eventgrad_amd.models import build_model
    from eventgrad_amd.ops import functional as O
    from eventgrad_amd.parallel.flat import FlatParamSpace
    from eventgrad_amd.ops.backend import native

    dev = torch.device("cuda")
    torch.manual_seed(0)
    m = build_model(name).to(dev)
    m.train()
    space = FlatParamSpace(m, dev)
    shape = (4, 1, 28, 28) if name in ("mlp", "cnn1", "cnn2") \
        else (4, 3, 32, 32)
    x = torch.randn(shape, device=dev)
    y = torch.randint(0, 10, (4,), device=dev)
    loss = O.nll_of_logits(m(x), y)
    loss.backward()
    native().sgd_step_norm(space.param, space.grad, space.momentum,
                           space.starts_t, space.numels_t, 0.01, 0.9, 0.0)
    torch.cuda.synchronize()
    assert torch.isfinite(loss).item(), name
    assert bool(torch.isfinite(space.grad).all().item()), name


def test_shadow_views_match_transforms():
    """FlatParamSpace's persistent bf16 KRSC/CRSK shadows must equal the
    per-use transform kernels' output, at init and after a param mutation
    + refresh_shadows()."""
    from eventgrad_amd.models import build_model
    from eventgrad_amd.models.layers import Conv2d, Linear
    from eventgrad_amd.ops.backend import native
    from eventgrad_amd.parallel.flat import FlatParamSpace

    torch.manual_seed(3)
    dev = torch.device("cuda")
    model = build_model("resnet20").to(dev)
    space = FlatParamSpace(model, dev)

    def check():
        n_conv = n_lin = 0
        for mod in model.modules():
            if isinstance(mod, Conv2d):
                wk_ref = native().oihw_to_krsc(mod.weight.detach().contiguous())
                assert torch.equal(mod._shadow_wk, wk_ref)
                wt_ref = native().krsc_to_crsk(wk_ref)
                assert torch.equal(mod._shadow_wt, wt_ref)
                n_conv += 1
            elif isinstance(mod, Linear):
                assert torch.equal(mod._shadow_wk,
                                   mod.weight.detach().to(torch.bfloat16))
                assert torch.equal(mod._shadow_wt, mod._shadow_wk.t())
                n_lin += 1
        assert n_conv > 0 and n_lin > 0

    check()
    with torch.no_grad():
        space.param.mul_(1.7).add_(0.01)
    space.refresh_shadows()
    torch.cuda.synchronize()
    check()


@pytest.mark.parametrize("model_name,mode", [("lenet5", "train"),
                                             ("resnet20", "eval")])
def test_direct_grads_match_fallback(model_name, mode):
    """Flat-space direct-grad writes (wgrad_into/bn outs/channel_sum_into)
    must equal the standalone allocate-and-return autograd path.

    Forward must be deterministic for a strict comparison, so BN models run
    in eval mode: train-mode BN batch stats are fp32 atomic reductions whose
    ordering differs run-to-run, flipping last bf16 bits that amplify over
    20 layers (measured ~20% per-tensor grad spread between two *identical*
    train-mode runs). Eval-BN still exercises the direct dgamma/dbeta and
    wgrad paths; lenet5 covers train mode (conv bias + linear, no BN).
    """
    from eventgrad_amd.models import build_model
    from eventgrad_amd.ops import functional as O
    from eventgrad_amd.parallel.flat import FlatParamSpace

    dev = torch.device("cuda")
    torch.manual_seed(5)
    m_ref = build_model(model_name).to(dev)     # standalone: fallback path
    torch.manual_seed(5)
    m_flat = build_model(model_name).to(dev)
    space = FlatParamSpace(m_flat, dev)         # direct path

    torch.manual_seed(6)
    x = torch.randn(16, 3, 32, 32, device=dev)
    y = torch.randint(0, 10, (16,), device=dev)

    from eventgrad_amd.models.layers import BatchNorm2d, Conv2d
    for m in (m_ref, m_flat):
        m.train()
        if mode == "eval":
            # deterministic forward: eval-mode BN (running stats) while the
            # model head stays in train mode (raw logits keep the graph
            # differentiable; eval log_softmax is a forward-only helper)
            for mod in m.modules():
                if isinstance(mod, (Conv2d, BatchNorm2d)):
                    mod.eval()
    torch.manual_seed(77)  # identical dropout seed draws in both runs
    out_r = m_ref(x)
    loss_r = O.nll_of_logits(out_r, y)
    loss_r.backward()
    space.zero_grad()
    torch.manual_seed(77)
    out_f = m_flat(x)
    loss_f = O.nll_of_logits(out_f, y)
    loss_f.backward()
    torch.cuda.synchronize()
    assert torch.equal(loss_r.detach(), loss_f.detach())
    for (name, pr), pf in zip(m_ref.named_parameters(), m_flat.parameters()):
        gr, gf = pr.grad, pf.grad
        denom = gr.norm().item() + 1e-12
        # both paths run the same fp32 atomic reductions; only summation
        # order differs
        err = (gr - gf).norm().item() / denom
        assert err < 1e-4, (name, err)


def test_bn_tail_fusion_grads_exact():
    """The bn_add_relu fused backward (relu_bwd_bnstats) must deposit
    exactly sum(da) / sum(da*xhat) into the BN's grad views, verified
    against fp32 math recomputed from the SAME run's saved stats."""
    from eventgrad_amd.models.resnet import BasicBlock
    from eventgrad_amd.parallel.flat import FlatParamSpace

    dev = torch.device("cuda")
    torch.manual_seed(9)
    block = BasicBlock(16, 16).to(dev)
    block.train()
    space = FlatParamSpace(block, dev)  # grad views -> fusion eligible

    x = torch.randn(8, 16, 16, 16, device=dev) \
        .permute(0, 2, 3, 1).contiguous().to(torch.bfloat16)
    space.zero_grad()
    y = block(x)
    y.float().sum().backward()      # dy of the block output == 1
    torch.cuda.synchronize()

    assert hasattr(block.bn2, "_bwd_stash"), "fusion did not engage"
    xb, mean, invstd = block.bn2._bwd_stash
    da = (y.detach().float() > 0).float()        # dy=1 gated by relu out
    xh = (xb.detach().float() - mean) * invstd
    C = xb.shape[-1]
    dbeta_ref = da.reshape(-1, C).sum(0)
    dgamma_ref = (da * xh).reshape(-1, C).sum(0)
    np.testing.assert_allclose(block.bn2.bias.grad.cpu().numpy(),
                               dbeta_ref.cpu().numpy(), rtol=1e-4, atol=1e-2)
    np.testing.assert_allclose(block.bn2.weight.grad.cpu().numpy(),
                               dgamma_ref.cpu().numpy(), rtol=1e-3, atol=5e-2)


def test_dgrad_bn_stats_fusion_exact():
    """bn1 -> conv2 fusion: the conv dgrad epilogue's accumulated BN stats
    must equal fp32 math on the retained upstream gradient."""
    from eventgrad_amd.models.resnet import BasicBlock
    from eventgrad_amd.ops import functional as O
    from eventgrad_amd.parallel.flat import FlatParamSpace

    dev = torch.device("cuda")
    torch.manual_seed(11)
    block = BasicBlock(64, 64).to(dev)
    block.train()
    space = FlatParamSpace(block, dev)

    x = torch.randn(4, 64, 8, 8, device=dev) \
        .permute(0, 2, 3, 1).contiguous().to(torch.bfloat16)  # NHWC
    # replicate the block body with a retained intermediate
    space.zero_grad()
    assert O.can_fuse_dgrad_stats(block.bn1, block.conv2, x)
    out1 = block.bn1(block.conv1(x, bn_stats=True), fuse_relu=True,
                     stats_ready=True, stats_consumer=True)
    out1.retain_grad()
    out2 = block.conv2(out1, bn_stats=True, dgrad_stats_bn=block.bn1)
    out = O.bn_add_relu(block.bn2, out2, x, stats_ready=True)
    out.float().sum().backward()
    torch.cuda.synchronize()

    dy1 = out1.grad.float()                      # conv2's dgrad output
    gated = torch.where(out1.detach().float() > 0, dy1,
                        torch.zeros(1, device=dev))
    x1, mean, invstd = block.bn1._bwd_stash
    xh = (x1.detach().float() - mean) * invstd
    C = 64
    dbeta_ref = gated.reshape(-1, C).sum(0)
    dgamma_ref = (gated * xh).reshape(-1, C).sum(0)
    np.testing.assert_allclose(block.bn1.bias.grad.cpu().numpy(),
                               dbeta_ref.cpu().numpy(), rtol=1e-3, atol=1e-2)
    np.testing.assert_allclose(block.bn1.weight.grad.cpu().numpy(),
                               dgamma_ref.cpu().numpy(), rtol=1e-3,
                               atol=5e-2)
