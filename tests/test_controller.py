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
