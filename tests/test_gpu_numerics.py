"""GPU numerics: every HIP kernel vs a plain PyTorch fp32 reference.

bf16-compute kernels are compared against fp32 torch with inputs
pre-quantized to bf16, using tolerances sized for bf16 rounding
(~8 mantissa bits -> relative ~1e-2 after short reductions).
"""

import numpy as np
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

DEV = "cuda"


def core():
    from eventgrad_amd.ops.backend import native
    return native()


def bq(x):
    """quantize to bf16 then back to fp32 (the oracle's input)."""
    return x.to(torch.bfloat16).float()


def rel_err(a, b):
    return ((a - b).norm() / (b.norm() + 1e-12)).item()


# ---------------------------------------------------------------- gemm ----

@pytest.mark.parametrize("M,N,K,bias,out_bf16", [
    (37, 53, 100, False, False),
    (64, 64, 64, True, True),
    (256, 10, 500, True, True),
    (128, 784, 256, False, False),
    (1, 10, 50, True, True),
])
def test_gemm_bias(M, N, K, bias, out_bf16):
    torch.manual_seed(0)
    A = torch.randn(M, K, device=DEV).to(torch.bfloat16)
    B = torch.randn(N, K, device=DEV).to(torch.bfloat16)
    b = torch.randn(N, device=DEV) if bias else torch.empty(0, device=DEV)
    C = core().gemm_bias(A, B, b, out_bf16)
    ref = bq(A.float()).cpu() @ bq(B.float()).cpu().t()
    if bias:
        ref = ref + b.cpu()
    assert rel_err(C.float().cpu(), ref) < 2e-2


# ---------------------------------------------------------------- conv ----

CONV_CASES = [
    # N, H, W, C, K, R, stride, pad, bias
    (2, 16, 16, 8, 16, 3, 1, 1, False),
    (2, 32, 32, 3, 64, 3, 1, 1, False),    # stem (C=3 slow path)
    (2, 16, 16, 64, 128, 3, 2, 1, False),  # strided stage transition
    (2, 16, 16, 64, 128, 1, 2, 0, False),  # 1x1 downsample
    (2, 28, 28, 1, 10, 3, 1, 0, True),     # CNN-2 conv1 (bias, no pad)
    (2, 13, 13, 10, 20, 3, 1, 0, True),    # CNN-2 conv2
    (2, 8, 8, 512, 512, 3, 1, 1, False),   # stage-4
]


@pytest.mark.parametrize("case", CONV_CASES)
def test_conv2d_autograd(case):
    N, H, W, C, K, R, stride, pad, bias = case
    torch.manual_seed(1)
    from eventgrad_amd.ops import functional as O

    x_f = torch.randn(N, C, H, W)
    w_f = torch.randn(K, C, R, R) * (1.0 / (R * np.sqrt(C)))
    b_f = torch.randn(K) if bias else None

    # GPU path (NHWC bf16)
    x_g = bq(x_f).to(DEV).permute(0, 2, 3, 1).contiguous().to(torch.bfloat16)
    x_g.requires_grad_(True)
    w_g = w_f.clone().to(DEV).requires_grad_(True)
    b_g = b_f.clone().to(DEV).requires_grad_(True) if bias else None
    y_g = O.conv2d(x_g, w_g, b_g, stride, pad)

    # CPU fp32 oracle on bf16-quantized inputs
    x_c = bq(x_f).requires_grad_(True)
    w_c = bq(w_f).requires_grad_(True)
    b_c = bq(b_f).requires_grad_(True) if bias else None
    y_c = F.conv2d(x_c, w_c, b_c, stride=stride, padding=pad)

    y_g_nchw = y_g.float().permute(0, 3, 1, 2).cpu()
    assert rel_err(y_g_nchw, y_c.detach()) < 3e-2, "fwd"

    dy = torch.randn_like(y_c)
    y_c.backward(dy)
    dy_g = bq(dy).to(DEV).permute(0, 2, 3, 1).contiguous().to(torch.bfloat16)
    y_g.backward(dy_g)

    assert rel_err(w_g.grad.cpu(), w_c.grad) < 3e-2, "wgrad"
    dx_g = x_g.grad.float().permute(0, 3, 1, 2).cpu()
    assert rel_err(dx_g, x_c.grad) < 3e-2, "dgrad"
    if bias:
        assert rel_err(b_g.grad.cpu(), b_c.grad) < 2e-2, "bias grad"


# ------------------------------------------------------------------ bn ----

@pytest.mark.parametrize("relu", [False, True])
def test_bn_autograd(relu):
    torch.manual_seed(2)
    from eventgrad_amd.ops import functional as O
    N, H, W, C = 4, 8, 8, 32
    x_f = torch.randn(N, C, H, W) * 2 + 1

    x_g = bq(x_f).to(DEV).permute(0, 2, 3, 1).contiguous().to(torch.bfloat16)
    x_g.requires_grad_(True)
    g_g = torch.rand(C, device=DEV).requires_grad_(True)
    b_g = torch.randn(C, device=DEV).requires_grad_(True)
    rm_g = torch.zeros(C, device=DEV)
    rv_g = torch.ones(C, device=DEV)
    y_g = O.batch_norm(x_g, g_g, b_g, rm_g, rv_g, True, 0.1, 1e-5, relu)

    x_c = bq(x_f).requires_grad_(True)
    g_c = g_g.detach().cpu().requires_grad_(True)
    b_c = b_g.detach().cpu().requires_grad_(True)
    rm_c = torch.zeros(C)
    rv_c = torch.ones(C)
    y_c = F.batch_norm(x_c, rm_c, rv_c, g_c, b_c, True, 0.1, 1e-5)
    if relu:
        y_c = F.relu(y_c)

    assert rel_err(y_g.float().permute(0, 3, 1, 2).cpu(), y_c.detach()) < 3e-2
    assert rel_err(rm_g.cpu(), rm_c) < 2e-2
    assert rel_err(rv_g.cpu(), rv_c) < 2e-2

    dy = torch.randn_like(y_c)
    y_c.backward(dy)
    y_g.backward(bq(dy).to(DEV).permute(0, 2, 3, 1).contiguous()
                 .to(torch.bfloat16))
    assert rel_err(g_g.grad.cpu(), g_c.grad) < 3e-2
    assert rel_err(b_g.grad.cpu(), b_c.grad) < 3e-2
    dx = x_g.grad.float().permute(0, 3, 1, 2).cpu()
    assert rel_err(dx, x_c.grad) < 4e-2


# ----------------------------------------------------- elementwise/pool ----

def test_relu_and_add_relu():
    from eventgrad_amd.ops import functional as O
    torch.manual_seed(3)
    a = torch.randn(3, 4, 4, 24, device=DEV).to(torch.bfloat16)
    b = torch.randn_like(a)
    a.requires_grad_(True)
    b.requires_grad_(True)
    y = O.add_relu(a, b)
    ref = F.relu(a.detach().float() + b.detach().float())
    assert rel_err(y.float().cpu(), ref.cpu()) < 2e-2
    dy = torch.randn_like(a)
    y.backward(dy)
    mask = (ref > 0).float()
    assert rel_err(a.grad.float().cpu(), (dy.float() * mask).cpu()) < 2e-2

    x = torch.randn(1000, device=DEV).to(torch.bfloat16).requires_grad_(True)
    z = O.relu(x)
    assert torch.equal(z.float().cpu(),
                       F.relu(x.detach().float()).cpu())


def test_pools():
    from eventgrad_amd.ops import functional as O
    torch.manual_seed(4)
    x_f = torch.randn(2, 8, 12, 12)
    xg = bq(x_f).to(DEV).permute(0, 2, 3, 1).contiguous().to(torch.bfloat16)
    xg.requires_grad_(True)
    y = O.max_pool2x2(xg)
    xc = bq(x_f).requires_grad_(True)
    yc = F.max_pool2d(xc, 2)
    assert rel_err(y.float().permute(0, 3, 1, 2).cpu(), yc.detach()) < 1e-2
    dy = torch.randn_like(yc)
    yc.backward(dy)
    y.backward(bq(dy).to(DEV).permute(0, 2, 3, 1).contiguous()
               .to(torch.bfloat16))
    assert rel_err(xg.grad.float().permute(0, 3, 1, 2).cpu(), xc.grad) < 2e-2

    x2 = torch.randn(2, 8, 8, 16)
    xg2 = bq(x2).to(DEV).permute(0, 2, 3, 1).contiguous().to(torch.bfloat16)
    xg2.requires_grad_(True)
    y2 = O.avg_pool(xg2, 4)
    xc2 = bq(x2).requires_grad_(True)
    yc2 = F.avg_pool2d(xc2, 4)
    assert rel_err(y2.float().permute(0, 3, 1, 2).cpu(), yc2.detach()) < 2e-2
    dy2 = torch.randn_like(yc2)
    yc2.backward(dy2)
    y2.backward(bq(dy2).to(DEV).permute(0, 2, 3, 1).contiguous()
                .to(torch.bfloat16))
    assert rel_err(xg2.grad.float().permute(0, 3, 1, 2).cpu(), xc2.grad) < 2e-2


def test_loss():
    from eventgrad_amd.ops import functional as O
    torch.manual_seed(5)
    logits = torch.randn(64, 10)
    tgt = torch.randint(0, 10, (64,))
    lg = bq(logits).to(DEV).to(torch.bfloat16).requires_grad_(True)
    tg = tgt.to(DEV)
    loss_g = O.nll_of_logits(lg, tg)
    lc = bq(logits).requires_grad_(True)
    loss_c = F.nll_loss(F.log_softmax(lc, 1), tgt)
    assert abs(loss_g.item() - loss_c.item()) < 2e-2
    loss_g.backward()
    loss_c.backward()
    assert rel_err(lg.grad.float().cpu(), lc.grad) < 2e-2


def test_dropout_stats():
    from eventgrad_amd.ops.backend import native
    x = torch.ones(100000, device=DEV).to(torch.bfloat16)
    y, mask = native().dropout_fwd(x, 0.5, 1234, False)
    keep = mask.float().mean().item()
    assert 0.47 < keep < 0.53
    # kept elements scaled by 1/(1-p)
    assert abs(y.float().sum().item() - 2.0 * mask.float().sum().item()) < 10
    # deterministic for same seed
    y2, mask2 = native().dropout_fwd(x, 0.5, 1234, False)
    assert torch.equal(mask, mask2)


# ------------------------------------------------------- engine kernels ----

def _mk_flat(sz=5, seed=0):
    torch.manual_seed(seed)
    numels = [100, 64, 1280, 7, 333][:sz]
    starts, off = [], 0
    for n in numels:
        starts.append(off)
        off += (n + 63) // 64 * 64
    total = off
    flat = torch.zeros(total, device=DEV)
    for s, n in zip(starts, numels):
        flat[s:s + n] = torch.randn(n, device=DEV)
    st = torch.tensor(starts, dtype=torch.int64, device=DEV)
    nu = torch.tensor(numels, dtype=torch.int64, device=DEV)
    return flat, st, nu, starts, numels


def test_seg_sqnorms():
    flat, st, nu, starts, numels = _mk_flat()
    out = core().seg_sqnorms(flat, st, nu)
    for i, (s, n) in enumerate(zip(starts, numels)):
        ref = flat[s:s + n].square().sum().item()
        assert abs(out[i].item() - ref) / (ref + 1e-9) < 1e-5


def test_sgd_step_norm_vs_cpu():
    flat, st, nu, starts, numels = _mk_flat(seed=7)
    grad = torch.randn_like(flat)
    # zero the pad gaps of grad like the real flat-grad buffer
    mask = torch.zeros_like(flat)
    for s, n in zip(starts, numels):
        mask[s:s + n] = 1
    grad *= mask
    mom = torch.randn_like(flat) * mask
    p_ref = flat.cpu().clone()
    g_ref = grad.cpu().clone()
    m_ref = mom.cpu().clone()
    norms = core().sgd_step_norm(flat, grad, mom, st, nu, 0.1, 0.9, 0.0)
    m_ref.mul_(0.9).add_(g_ref)
    p_ref.add_(m_ref, alpha=-0.1)
    assert rel_err(flat.cpu(), p_ref) < 1e-6
    assert rel_err(mom.cpu(), m_ref) < 1e-6
    for i, (s, n) in enumerate(zip(starts, numels)):
        ref = p_ref[s:s + n].square().sum().item()
        assert abs(norms[i].item() - ref) / (ref + 1e-9) < 1e-5


def test_avg3():
    flat, *_ = _mk_flat(seed=8)
    l = torch.randn_like(flat)
    r = torch.randn_like(flat)
    ref = (flat + l + r) / 3
    core().avg3(flat, l, r)
    assert rel_err(flat, ref) < 1e-6


def test_gather_scatter_segments():
    flat, st, nu, starts, numels = _mk_flat(seed=9)
    fired = [0, 2, 4]
    src = torch.tensor([starts[i] for i in fired], device=DEV)
    sizes = [numels[i] for i in fired]
    cum = np.concatenate([[0], np.cumsum(sizes)])
    offs = torch.tensor(cum, dtype=torch.int64, device=DEV)
    payload = core().gather_segments(flat, src, offs, int(cum[-1]))
    ref = torch.cat([flat[starts[i]:starts[i] + numels[i]] for i in fired])
    assert torch.equal(payload, ref)
    inbox = torch.zeros_like(flat)
    core().scatter_segments(payload, src, offs, inbox)
    for i in fired:
        assert torch.equal(inbox[starts[i]:starts[i] + numels[i]],
                           flat[starts[i]:starts[i] + numels[i]])


def test_trigger_update_matches_host_controller():
    from eventgrad_amd.parallel.controller import TriggerController
    sz, H = 8, 2
    host = TriggerController(sz, adaptive=True, horizon=1.05, constant=5e-4,
                             sent_history=H, initial_comm_passes=3)
    thres = torch.zeros(sz, device=DEV)
    lsn = torch.zeros(sz, device=DEV)
    lsi = torch.zeros(sz, device=DEV)
    slopes = torch.zeros(sz * H, device=DEV)
    nev = torch.zeros(1, dtype=torch.int32, device=DEV)
    rng = np.random.default_rng(0)
    for p in range(1, 30):
        norms = rng.random(sz).astype(np.float32) * (1 + p / 10)
        mh = host.step(norms, p)
        nd = torch.tensor(norms**2, device=DEV)
        mg = core().trigger_update(nd, thres, lsn, lsi, slopes, nev, p, True,
                                   1.05, 5e-4, 3, False)
        assert (mg.cpu().numpy().astype(bool) == mh).all(), p
        np.testing.assert_allclose(thres.cpu().numpy(), host.thres,
                                   rtol=1e-4, atol=1e-7)
    assert int(nev.item()) == host.num_events


def test_trigger_decide_matches_update_mask():
    """The pure decide kernel (lookahead mask posting) must produce exactly
    the mask trigger_update commits from the same state."""
    sz, H = 8, 2
    thres = torch.zeros(sz, device=DEV)
    lsn = torch.zeros(sz, device=DEV)
    lsi = torch.zeros(sz, device=DEV)
    slopes = torch.zeros(sz * H, device=DEV)
    nev = torch.zeros(1, dtype=torch.int32, device=DEV)
    rng = np.random.default_rng(7)
    for p in range(1, 40):
        nd = torch.tensor((rng.random(sz).astype(np.float32)
                           * (1 + p / 10)) ** 2, device=DEV)
        md = core().trigger_decide(nd, thres, lsn, p, True, 1.05, 5e-4, 3)
        mu = core().trigger_update(nd, thres, lsn, lsi, slopes, nev, p, True,
                                   1.05, 5e-4, 3, False)
        assert torch.equal(md, mu), p


def test_topk_absdiff():
    torch.manual_seed(11)
    n, k = 10000, 77
    x = torch.randn(n, device=DEV)
    prev = torch.randn(n, device=DEV)
    prev_copy = prev.clone()
    vals, idx = core().topk_absdiff(x, prev, k)
    diff = (x - prev_copy).abs()
    ref_vals, ref_idx = torch.topk(diff, k)
    # same selected |diff| multiset (ties may reorder)
    got = diff[idx.long()].sort().values
    assert torch.allclose(got, ref_vals.sort().values, atol=1e-6)
    # vals are x at the selected indices; prev updated there
    assert torch.allclose(vals, x[idx.long()])
    assert torch.allclose(prev[idx.long()], x[idx.long()])
    untouched = torch.ones(n, dtype=torch.bool, device=DEV)
    untouched[idx.long()] = False
    assert torch.equal(prev[untouched], prev_copy[untouched])


def test_conv_fused_bn_stats_matches_separate():
    """conv(bn_stats=True) + bn_fwd(have_stats) == conv + separate stats."""
    torch.manual_seed(21)
    from eventgrad_amd.ops import functional as O
    N, H, W, C, K = 4, 16, 16, 32, 64
    x = torch.randn(N, H, W, C, device=DEV).to(torch.bfloat16)
    w = torch.randn(K, C, 3, 3, device=DEV) * 0.05
    gamma = torch.rand(K, device=DEV)
    beta = torch.randn(K, device=DEV)

    def run(fused):
        rm = torch.zeros(K, device=DEV)
        rv = torch.ones(K, device=DEV)
        y = O.conv2d(x.clone(), w, None, 1, 1, bn_stats=fused)
        out = O.batch_norm(y, gamma, beta, rm, rv, True, 0.1, 1e-5,
                           fuse_relu=True, stats_ready=fused)
        return out.float(), rm.clone(), rv.clone()

    o1, rm1, rv1 = run(False)
    o2, rm2, rv2 = run(True)
    assert rel_err(o2, o1) < 1e-2
    # fused stats sum the fp32 pre-rounding conv outputs, the separate pass
    # re-reads the bf16-rounded y — the fused values are the more accurate;
    # means are near zero so compare absolutely at bf16-noise scale
    assert torch.allclose(rm2, rm1, atol=3e-4), (rm2 - rm1).abs().max()
    assert rel_err(rv2, rv1) < 1e-2


def test_spevent_pack_unpack_batched():
    """Fused multi-tensor pack/unpack vs the single-tensor kernels."""
    torch.manual_seed(33)
    dev = "cuda"
    # three segments in one flat buffer, 64-aligned starts
    numels = [1000, 257, 4096]
    starts, off = [], 0
    for n in numels:
        starts.append(off)
        off += (n + 63) // 64 * 64
    flat = torch.zeros(off, device=dev)
    prev = torch.zeros(off, device=dev)
    for s, n in zip(starts, numels):
        flat[s:s + n] = torch.randn(n, device=dev)
        prev[s:s + n] = torch.randn(n, device=dev)
    prev0 = prev.clone()
    ks = [50, 17, 123]
    val_offs, cum_k, o, ck = [], [], 0, 0
    for k in ks:
        val_offs.append(o)
        cum_k.append(ck)
        o += 2 * k
        ck += k
    t64 = lambda v: torch.tensor(v, dtype=torch.int64, device=dev)
    payload = core().spevent_pack(flat, prev, t64(starts), t64(numels),
                                  t64(ks), t64(val_offs + [o] + cum_k + [ck]),
                                  o, max(numels))
    # reference: per-segment torch.topk of |flat - prev0|
    off2 = 0
    for (s, n, k) in zip(starts, numels, ks):
        seg, pseg = flat[s:s + n], prev0[s:s + n]
        ref_vals, _ = torch.topk((seg - pseg).abs(), k)
        got_vals = payload[off2:off2 + k]
        got_idx = payload[off2 + k:off2 + 2 * k].view(torch.int32).long()
        # same selected |diff| multiset; vals are flat at the indices
        got_diff = (seg - pseg).abs()[got_idx].sort().values
        assert torch.allclose(got_diff, ref_vals.sort().values, atol=1e-6)
        assert torch.allclose(got_vals, seg[got_idx])
        # prev updated exactly at the selected indices
        assert torch.allclose(prev[s:s + n][got_idx], seg[got_idx])
        mask = torch.ones(n, dtype=torch.bool, device=dev)
        mask[got_idx] = False
        assert torch.equal(prev[s:s + n][mask], pseg[mask])
        off2 += 2 * k
    # unpack scatters into a replica exactly like per-segment scatter
    replica = torch.randn(off, device=dev)
    ref_replica = replica.clone()
    core().spevent_unpack(payload, t64(starts), t64(ks),
                          t64(val_offs + [o] + cum_k + [ck]), replica, ck)
    off2 = 0
    for (s, n, k) in zip(starts, numels, ks):
        vals = payload[off2:off2 + k]
        idx = payload[off2 + k:off2 + 2 * k].view(torch.int32).long()
        ref_replica[s:s + n][idx] = vals
        off2 += 2 * k
    assert torch.equal(replica, ref_replica)


def test_conv2d_fuzz_shapes():
    """Seeded fuzz: random conv geometries vs the fp32 oracle."""
    rng = np.random.default_rng(7)
    from eventgrad_amd.ops import functional as O
    for trial in range(10):
        N = int(rng.integers(1, 5))
        C = int(rng.choice([3, 8, 16, 24, 40, 64, 96]))
        K = int(rng.choice([8, 16, 32, 48, 64, 128]))
        R = int(rng.choice([1, 3, 5]))
        H = int(rng.integers(R, 20))
        stride = int(rng.choice([1, 2]))
        pad = int(rng.integers(0, (R + 1) // 2 + 1))
        if (H + 2 * pad - R) // stride + 1 < 1:
            continue
        torch.manual_seed(100 + trial)
        x_f = torch.randn(N, C, H, H)
        w_f = torch.randn(K, C, R, R) * (0.5 / (R * np.sqrt(C)))
        x_g = bq(x_f).to(DEV).permute(0, 2, 3, 1).contiguous() \
            .to(torch.bfloat16).requires_grad_(True)
        w_g = w_f.clone().to(DEV).requires_grad_(True)
        y_g = O.conv2d(x_g, w_g, None, stride, pad)
        x_c = bq(x_f).requires_grad_(True)
        w_c = bq(w_f).requires_grad_(True)
        y_c = F.conv2d(x_c, w_c, None, stride=stride, padding=pad)
        tag = f"trial{trial} N{N} C{C} K{K} R{R} H{H} s{stride} p{pad}"
        assert rel_err(y_g.float().permute(0, 3, 1, 2).cpu(),
                       y_c.detach()) < 4e-2, ("fwd", tag)
        dy = torch.randn_like(y_c)
        y_c.backward(dy)
        y_g.backward(bq(dy).to(DEV).permute(0, 2, 3, 1).contiguous()
                     .to(torch.bfloat16))
        assert rel_err(w_g.grad.cpu(), w_c.grad) < 4e-2, ("wgrad", tag)
        assert rel_err(x_g.grad.float().permute(0, 3, 1, 2).cpu(),
                       x_c.grad) < 4e-2, ("dgrad", tag)


def test_gemm_fuzz_shapes():
    rng = np.random.default_rng(11)
    for trial in range(10):
        M = int(rng.integers(1, 300))
        N = int(rng.integers(1, 300))
        K = int(rng.integers(1, 600))
        torch.manual_seed(200 + trial)
        A = torch.randn(M, K, device=DEV).to(torch.bfloat16)
        B = torch.randn(N, K, device=DEV).to(torch.bfloat16)
        C = core().gemm_bias(A, B, torch.empty(0, device=DEV), False)
        ref = bq(A.float()).cpu() @ bq(B.float()).cpu().t()
        assert rel_err(C.cpu(), ref) < 3e-2, (trial, M, N, K)


def test_sgd_step_norm_weight_decay():
    flat, st, nu, starts, numels = _mk_flat(seed=17)
    mask = torch.zeros_like(flat)
    for s, n in zip(starts, numels):
        mask[s:s + n] = 1
    grad = torch.randn_like(flat) * mask
    mom = torch.randn_like(flat) * mask
    p_ref, g_ref, m_ref = flat.cpu().clone(), grad.cpu(), mom.cpu().clone()
    core().sgd_step_norm(flat, grad, mom, st, nu, 0.05, 0.9, 1e-4)
    g_eff = g_ref + 1e-4 * p_ref
    m_ref.mul_(0.9).add_(g_eff)
    p_ref.add_(m_ref, alpha=-0.05)
    # pad gaps get wd*0 contributions only where param is 0 -> identical
    assert rel_err(flat.cpu() * mask.cpu(), p_ref * mask.cpu()) < 1e-6
    assert rel_err(mom.cpu() * mask.cpu(), m_ref * mask.cpu()) < 1e-6


def test_relu_bwd_bnstats_fused():
    """Fused add_relu backward + producing-BN stats == the separate
    relu_bwd + sum(da), sum(da*xhat) computed in fp32 torch."""
    torch.manual_seed(4)
    rows, c = 1024, 64
    dy = torch.randn(rows, c, device=DEV).to(torch.bfloat16)
    y = torch.randn(rows, c, device=DEV).to(torch.bfloat16)   # relu out sign
    xb = torch.randn(rows, c, device=DEV).to(torch.bfloat16)
    mean = torch.randn(c, device=DEV)
    invstd = torch.rand(c, device=DEV) + 0.5
    dgamma = torch.zeros(c, device=DEV)
    dbeta = torch.zeros(c, device=DEV)
    da = core().relu_bwd_bnstats(dy, y, xb, mean, invstd, dgamma, dbeta)
    da_ref = torch.where(y.float() > 0, dy.float(), torch.zeros(1, device=DEV))
    assert torch.equal(da.float(), da_ref.to(torch.bfloat16).float())
    xh = (xb.float() - mean) * invstd
    np.testing.assert_allclose(dbeta.cpu().numpy(),
                               da_ref.sum(0).cpu().numpy(), rtol=1e-4,
                               atol=1e-3)
    np.testing.assert_allclose(dgamma.cpu().numpy(),
                               (da_ref * xh).sum(0).cpu().numpy(), rtol=1e-3,
                               atol=1e-2)
    # accumulate semantics: second call adds
    core().relu_bwd_bnstats(dy, y, xb, mean, invstd, dgamma, dbeta)
    np.testing.assert_allclose(dbeta.cpu().numpy(),
                               2 * da_ref.sum(0).cpu().numpy(), rtol=1e-4,
                               atol=2e-3)
