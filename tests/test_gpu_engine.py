"""GPU gossip-engine integration via a loopback transport.

Multi-GPU runs can't happen on the 1-GPU test box, so this exercises the
exact GPU code path of GossipEngine/SparseGossipEngine (device-resident
trigger controller, payload gather/scatter kernels, avg3) with a fake
ring where both neighbors are this rank itself. Self-ring invariant:
averaging with two copies of yourself leaves parameters unchanged.
"""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


class LoopbackTransport:
    """Both neighbors are self: payloads come straight back."""

    def __init__(self):
        self._sent = None
        self._mask = None

    def exchange_masks(self, mask):
        m = mask.cpu()
        return m.clone(), m.clone()

    # lookahead protocol (engine.step posts next pass's mask)
    def post_masks(self, mask):
        self._mask = mask.cpu()

    def wait_masks(self):
        m, self._mask = self._mask, None
        return m.clone(), m.clone()

    def cancel_pending_masks(self):
        self._mask = None

    def post_payloads(self, send_l, send_r, recv_l, recv_r):
        if send_l is not None and recv_l is not None and send_l.numel():
            recv_l.copy_(send_l)
        if send_r is not None and recv_r is not None and send_r.numel():
            recv_r.copy_(send_r)

    def finish(self):
        pass


def _engine(mode, model_name="cnn2"):
    from eventgrad_amd.config import RunConfig, TriggerConfig, OptimConfig
    from eventgrad_amd.models import build_model
    from eventgrad_amd.parallel.engine import (GossipEngine,
                                               SparseGossipEngine)
    from eventgrad_amd.parallel.flat import FlatParamSpace

    dev = torch.device("cuda")
    torch.manual_seed(0)
    model = build_model(model_name).to(dev)
    space = FlatParamSpace(model, dev)
    cfg = RunConfig(mode=mode, optim=OptimConfig(lr=0.05, momentum=0.9),
                    trigger=TriggerConfig(adaptive=True, horizon=1.05,
                                          initial_comm_passes=3))
    cfg.topk_percent = 5.0
    cls = SparseGossipEngine if mode == "spevent" else GossipEngine
    eng = cls(space, cfg, rank=0, world=2, device=dev)
    eng.transport = LoopbackTransport()
    return model, space, eng


@pytest.mark.parametrize("mode", ["decent", "event", "spevent"])
def test_loopback_self_average_identity(mode):
    model, space, eng = _engine(mode)
    before = space.clone_params()
    eng.begin_pass(1)      # warmup/always -> every tensor fires
    eng.after_backward()   # inbox == own params; (p+p+p)/3 == p
    assert torch.allclose(space.param, before, rtol=1e-6, atol=1e-7)
    assert eng.num_events == 2 * space.sz


def test_event_engine_device_controller_dynamics():
    """Full GPU event-engine loop: warmup fires everything, the adaptive
    threshold state evolves to finite positive values, and post-warmup
    passes skip at least some sends (kernel<->host controller parity is
    covered bit-level in test_gpu_numerics)."""
    from eventgrad_amd.ops import functional as O

    model, space, eng = _engine("event")
    x = torch.randn(16, 1, 28, 28, device="cuda")
    y = torch.randint(0, 10, (16,), device="cuda")
    events_at_warmup_end = None
    for p in range(1, 16):
        eng.begin_pass(p)
        if p < 3:  # initial_comm_passes = 3
            assert eng.ctrl.last_fired.all(), p
        space.zero_grad()
        loss = O.nll_of_logits(model(x), y)
        loss.backward()
        eng.after_backward()
        eng.step()
        if p == 2:
            events_at_warmup_end = eng.num_events
    thres = eng.ctrl.thres.cpu().numpy()
    assert np.isfinite(thres).all() and (thres > 0).all()
    post = eng.num_events - events_at_warmup_end
    assert 0 <= post < 2 * space.sz * 13  # some sends skipped post-warmup


def test_spevent_loopback_replicas_track_params():
    model, space, eng = _engine("spevent")
    x = torch.randn(8, 1, 28, 28, device="cuda")
    y = torch.randint(0, 10, (8,), device="cuda")
    from eventgrad_amd.ops import functional as O

    for p in range(1, 6):
        eng.begin_pass(p)
        space.zero_grad()
        O.nll_of_logits(model(x), y).backward()
        eng.after_backward()
        eng.step()
    # replicas received top-k updates of our own params each fired pass;
    # they must stay finite and not all-zero, and prev must differ from 0
    assert torch.isfinite(eng.inbox_left).all()
    assert eng.inbox_left.abs().sum() > 0
    assert torch.isfinite(eng.prev).all()
