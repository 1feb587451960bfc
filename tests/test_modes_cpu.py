"""Multi-process (gloo, world 2) end-to-end tests of all four modes.

These exercise the exact code path used on GPU (same engines/transport; the
backend string and math kernels differ), per SURVEY.md §4 strategy (c).
"""

import glob
import json
import os

import pytest
import torch

from util_dist import run_world
import dist_workers as W


@pytest.mark.parametrize("mode", ["cent", "decent", "event", "spevent"])
def test_mode_runs_world2(tmp_path, mode):
    run_world(W.train_mode_worker, 2, mode, str(tmp_path))
    files = sorted(glob.glob(os.path.join(tmp_path, f"{mode}_r*.pt")))
    assert len(files) == 2
    outs = [torch.load(f, weights_only=False) for f in files]
    for o in outs:
        assert torch.isfinite(o["param"]).all()
    # after the closing consensus allreduce all ranks hold the same model
    assert torch.allclose(outs[0]["param"], outs[1]["param"], atol=1e-6)
    m = outs[0]["metrics"]
    assert m["total_passes"] == 8  # 2 epochs x (256/2 shard / 32 batch)
    if mode == "decent":
        assert m["messages_saved_pct"] == 0.0
    if mode == "event":
        # 5-pass warmup fires all; afterwards some tensors skip
        assert 0.0 <= m["messages_saved_pct"] < 100.0


def test_event_thres0_identical_to_decent(tmp_path):
    """The built-in A/B control (dmnist/event/README.md:59-60)."""
    run_world(W.event_equals_decent_worker, 2, str(tmp_path))
    for f in glob.glob(os.path.join(tmp_path, "eqdec_r*.pt")):
        assert torch.load(f, weights_only=False)["identical"]


def test_event_thres0_identical_to_decent_world4(tmp_path):
    """World-4 ring wiring check: event(thres=0, no warmup) runs the full
    dynamic mask-exchange + matched-payload protocol with every tensor
    firing, and must be bitwise-identical to decent's static fast path
    (de-risk the >2-rank wire)."""
    run_world(W.event_equals_decent_worker, 4, str(tmp_path))
    files = glob.glob(os.path.join(tmp_path, "eqdec_r*.pt"))
    assert len(files) == 4
    for f in files:
        assert torch.load(f, weights_only=False)["identical"]


@pytest.mark.parametrize("mode", ["event", "spevent"])
def test_mode_runs_world4(tmp_path, mode):
    """Post-warmup sparse masks on a 4-rank ring (distinct left/right
    neighbors, unlike world 2 where both are the same peer)."""
    run_world(W.train_mode_worker, 4, mode, str(tmp_path))
    files = sorted(glob.glob(os.path.join(tmp_path, f"{mode}_r*.pt")))
    assert len(files) == 4
    outs = [torch.load(f, weights_only=False) for f in files]
    for o in outs:
        assert torch.isfinite(o["param"]).all()
        assert torch.allclose(outs[0]["param"], o["param"], atol=1e-6)
    m = outs[0]["metrics"]
    assert 0.0 <= m["messages_saved_pct"] < 100.0


def test_event_runs_world8(tmp_path):
    """8-rank ring — the BASELINE scaling config's world size, on the gloo
    wire (same engines/transport as the GPU nccl path)."""
    run_world(W.train_mode_worker, 8, "event", str(tmp_path),
              '{"batch_size": 16, "n_train": 256, "epochs": 2}')
    files = sorted(glob.glob(os.path.join(tmp_path, "event_r*.pt")))
    assert len(files) == 8
    outs = [torch.load(f, weights_only=False) for f in files]
    for o in outs:
        assert torch.isfinite(o["param"]).all()
        assert torch.allclose(outs[0]["param"], o["param"], atol=1e-6)
    assert 0.0 <= outs[0]["metrics"]["messages_saved_pct"] < 100.0


def test_event_deterministic_world4(tmp_path):
    """Two identical post-warmup event runs must match bit-for-bit: the
    two-phase mask+payload protocol is deterministic by construction."""
    run_world(W.event_twice_deterministic_worker, 4, str(tmp_path))
    for f in glob.glob(os.path.join(tmp_path, "det_r*.pt")):
        d = torch.load(f, weights_only=False)
        assert d["identical"]
        assert d["events0"] == d["events1"]


def test_cent_world2_deterministic(tmp_path):
    run_world(W.cent_equals_fullbatch_worker, 2, str(tmp_path))
    outs = [torch.load(f, weights_only=False) for f in
            sorted(glob.glob(os.path.join(tmp_path, "cent_r*.pt")))]
    assert torch.allclose(outs[0]["param"], outs[1]["param"], atol=1e-6)


def test_checkpoint_resume_identity(tmp_path):
    run_world(W.checkpoint_resume_worker, 2, str(tmp_path))
    for f in glob.glob(os.path.join(tmp_path, "ckres_r*.pt")):
        d = torch.load(f, weights_only=False)
        assert torch.allclose(d["ref"], d["resumed"], atol=1e-6), \
            (d["ref"] - d["resumed"]).abs().max()


def test_checkpoint_resume_bn_model(tmp_path):
    """BN running stats (buffers outside FlatParamSpace) must survive the
    checkpoint round-trip: params bit-identical AND eval identical."""
    run_world(W.checkpoint_resume_bn_worker, 2, str(tmp_path))
    files = sorted(glob.glob(os.path.join(tmp_path, "ckbn_r*.pt")))
    assert len(files) == 2
    for f in files:
        d = torch.load(f, weights_only=False)
        assert torch.allclose(d["ref"], d["resumed"], atol=1e-6)
        assert d["n_bufs"] > 0 and d["bufs_equal"]
        if d["ref_eval"] is not None:  # rank 0 evaluated
            assert d["res_eval"] == d["ref_eval"]
            assert abs(d["res_loss"] - d["ref_loss"]) < 1e-6


def test_event_runs_world3_odd_ring(tmp_path):
    """Odd ring: left != right for every rank and the ring is not
    edge-symmetric — exercises mask/payload pairing beyond the
    world-2 same-peer case and the even-ring world 4/8 tests."""
    run_world(W.train_mode_worker, 3, "event", str(tmp_path),
              '{"batch_size": 16, "n_train": 192}')
    files = sorted(glob.glob(os.path.join(tmp_path, "event_r*.pt")))
    assert len(files) == 3
    outs = [torch.load(f, weights_only=False) for f in files]
    for o in outs:
        assert torch.isfinite(o["param"]).all()
        assert torch.allclose(outs[0]["param"], o["param"], atol=1e-6)
    assert 0.0 <= outs[0]["metrics"]["messages_saved_pct"] < 100.0
