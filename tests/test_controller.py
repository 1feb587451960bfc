"""Trigger/adaptive-threshold controller semantics (event.cpp:324-392)."""

import numpy as np

from eventgrad_amd.parallel.controller import TriggerController


def make(sz=3, adaptive=True, horizon=1.01, constant=5e-4, warmup=3,
         always=False):
    return TriggerController(sz, adaptive=adaptive, horizon=horizon,
                             constant=constant, sent_history=2,
                             initial_comm_passes=warmup, always_fire=always)


def test_warmup_always_fires():
    c = make(warmup=5, adaptive=False, constant=1e9)
    for p in range(1, 5):
        fire = c.step(np.zeros(3, np.float32), p)
        assert fire.all()
    fire = c.step(np.zeros(3, np.float32), 5)
    assert not fire.any()  # past warmup, diff 0 < huge threshold


def test_constant_threshold_fire_rule():
    c = make(adaptive=False, constant=0.5, warmup=0)
    # pass 1: norms [0.6, 0.3]: diff vs last_sent(0) -> fire only tensor 0
    fire = c.step(np.array([0.6, 0.3, 0.0], np.float32), 1)
    assert fire.tolist() == [True, False, False]
    assert c.last_sent_norm[0] == np.float32(0.6)
    assert c.last_sent_norm[1] == 0.0  # not updated when unfired
    assert c.num_events == 2


def test_zero_threshold_always_fires():
    c = make(adaptive=False, constant=0.0, warmup=0)
    for p in range(1, 10):
        assert c.step(np.random.rand(3).astype(np.float32), p).all()


def test_adaptive_threshold_evolution():
    """Hand-computed: slopes average becomes the new threshold on fire."""
    c = make(sz=1, adaptive=True, horizon=2.0, warmup=0)
    # pass 1: thres = 0*2 = 0; diff = 1.0 >= 0 -> fire.
    # slope = 1.0/1; slopes=[0,1]; thres = 0.5
    fire = c.step(np.array([1.0], np.float32), 1)
    assert fire[0] and c.thres[0] == np.float32(0.5)
    # pass 2: thres pre-update = 1.0; diff = |1.2-1.0| = 0.2 < 1.0 -> no fire
    fire = c.step(np.array([1.2], np.float32), 2)
    assert not fire[0] and c.thres[0] == np.float32(1.0)
    # pass 3: thres = 2.0; diff = |3.5-1.0| = 2.5 >= 2 -> fire;
    # iter_diff = 3-1 = 2, slope = 1.25; slopes [1.0, 1.25] -> thres 1.125
    fire = c.step(np.array([3.5], np.float32), 3)
    assert fire[0]
    assert abs(c.thres[0] - 1.125) < 1e-6
    assert c.last_sent_iter[0] == 3.0
    assert c.num_events == 4


def test_always_fire_mode():
    c = make(always=True, warmup=0)
    for p in range(1, 5):
        assert c.step(np.random.rand(3).astype(np.float32), p).all()
    assert c.num_events == 2 * 3 * 4


def test_state_roundtrip():
    c = make()
    for p in range(1, 8):
        c.step(np.random.rand(3).astype(np.float32), p)
    d = c.state_dict()
    c2 = make()
    c2.load_state_dict(d)
    m1 = c.step(np.ones(3, np.float32), 8)
    m2 = c2.step(np.ones(3, np.float32), 8)
    assert (m1 == m2).all()
    assert np.array_equal(c.thres, c2.thres)


def test_adaptive_horizon_zero_always_fires():
    """dmnist/event/README.md:59-60: horizon 0 (adaptive) == the plain
    Lian et al. ring — the pre-update thres*0 wipes the threshold every
    pass even though firing writes a nonzero slope average."""
    c = make(adaptive=True, horizon=0.0, warmup=0)
    for p in range(1, 12):
        fire = c.step(np.random.rand(3).astype(np.float32) * 10, p)
        assert fire.all(), p


def test_decide_matches_step_mask():
    """decide() (pure lookahead) must equal the mask step() then commits,
    from any reachable state."""
    import numpy as np
    from eventgrad_amd.parallel.controller import TriggerController

    rng = np.random.default_rng(3)
    for adaptive in (True, False):
        ctrl = TriggerController(6, adaptive=adaptive, horizon=1.03,
                                 constant=5e-4, initial_comm_passes=4)
        for p in range(1, 50):
            norms = rng.random(6).astype(np.float32) * (1 + p / 20)
            d = ctrl.decide(norms, p)
            f = ctrl.step(norms, p)
            assert (d == f).all(), (adaptive, p)


def test_engine_mask_lookahead_protocol():
    """engine.step() must post the NEXT pass's mask exchange (decide-only,
    no state commit) and begin_pass must commit + wait — with the commit
    mask identical to the posted one, and no posts during warmup/decent
    (static all-fire)."""
    import numpy as np
    import torch
    from eventgrad_amd.config import RunConfig, TriggerConfig, OptimConfig
    from eventgrad_amd.models import build_model
    from eventgrad_amd.parallel.engine import GossipEngine
    from eventgrad_amd.parallel.flat import FlatParamSpace

    class RecordingTransport:
        def __init__(self):
            self.log = []
            self._mask = None

        def post_masks(self, mask):
            self.log.append(("post", mask.clone()))
            self._mask = mask.cpu()

        def wait_masks(self):
            self.log.append(("wait",))
            m, self._mask = self._mask, None
            return m.clone(), m.clone()

        def exchange_masks(self, mask):
            self.log.append(("blocking", mask.clone()))
            m = mask.cpu()
            return m.clone(), m.clone()

        def post_payloads(self, *a):
            self.log.append(("payloads",))

        def finish(self):
            pass

        def cancel_pending_masks(self):
            self.log.append(("cancel",))

    torch.manual_seed(0)
    model = build_model("mlp")
    dev = torch.device("cpu")
    space = FlatParamSpace(model, dev)
    cfg = RunConfig(mode="event", device="cpu",
                    optim=OptimConfig(lr=0.05),
                    trigger=TriggerConfig(adaptive=True, horizon=1.01,
                                          initial_comm_passes=3))
    eng = GossipEngine(space, cfg, rank=0, world=2, device=dev)
    tr = RecordingTransport()
    eng.transport = tr

    x = torch.randn(8, 1, 28, 28)
    y = torch.randint(0, 10, (8,))
    from eventgrad_amd.ops import functional as O
    for p in range(1, 8):
        eng.begin_pass(p)
        space.zero_grad()
        loss = O.nll_of_logits(model(x), y)
        loss.backward()
        eng.after_backward()
        eng.step()

    ops = [e[0] for e in tr.log]
    # passes 1-2 are warmup (static all-fire): no mask traffic at all
    first_post = ops.index("post")
    assert "blocking" not in ops[:first_post]
    # lookahead posted at step(p) for p+1 >= warmup; begin_pass waits.
    # The LAST step's post is for a pass that never runs — finalize drains
    # it (cancel path), so posts == waits + 1 with a pending decision left.
    assert eng._pending is not None
    assert ops.count("post") == ops.count("wait") + 1
    # strict alternation post -> wait (never two posts in flight)
    seq = [o for o in ops if o in ("post", "wait")]
    assert all(a == "post" and b == "wait"
               for a, b in zip(seq[::2], seq[1::2]))
    eng.transport.cancel_pending_masks()
    assert tr.log[-1][0] == "cancel"
