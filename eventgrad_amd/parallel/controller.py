"""EventGraD trigger + adaptive-threshold controller (host mirror).

Semantics transcribed from dmnist/event/event.cpp:324-392 (identical in
dcifar10/event/event.cpp:299-365 and spevent.cpp:321-426):

  per pass, per tensor i (norm = ||theta_i||_2 of the CURRENT parameter,
  i.e. the value produced by the previous optimizer step, before this
  pass's neighbor averaging):

    value_diff = |norm_i - last_sent_norm[i]|
    iter_diff  = pass_num - last_sent_iter[i]
    thres[i]   = thres[i] * horizon      (adaptive)   # pre-update
               = constant                (static)
    fire_i     = value_diff >= thres[i]  or  pass_num < initial_comm_passes
    if fire_i:
        slopes[i] <- shift left; slopes[i][-1] = value_diff / iter_diff
        if adaptive: thres[i] = mean(slopes[i])
        last_sent_norm[i] = norm_i ; last_sent_iter[i] = pass_num
        num_events += 2                  # one per neighbor

The controller state is tiny (sz <= 86 scalars per array) — on GPU the same
update runs device-resident in the HIP trigger kernel (csrc/engine.hip trigger_update)
and this class is used for CPU runs, unit tests, and kernel parity checks.
"""

from __future__ import annotations

import numpy as np


class TriggerController:
    def __init__(self, sz: int, adaptive: bool, horizon: float,
                 constant: float, sent_history: int = 2,
                 initial_comm_passes: int = 30,
                 always_fire: bool = False):
        self.sz = sz
        self.adaptive = adaptive
        self.horizon = np.float32(horizon)
        self.constant = np.float32(constant)
        self.sent_history = sent_history
        self.initial_comm_passes = initial_comm_passes
        self.always_fire = always_fire

        self.thres = np.zeros(sz, dtype=np.float32)
        self.last_sent_norm = np.zeros(sz, dtype=np.float32)
        self.last_sent_iter = np.zeros(sz, dtype=np.float32)
        self.slopes = np.zeros((sz, sent_history), dtype=np.float32)
        self.num_events = 0
        # last evaluated values, for trace files
        self.last_norms = np.zeros(sz, dtype=np.float32)
        self.last_fired = np.zeros(sz, dtype=bool)

    def decide(self, norms: np.ndarray, pass_num: int) -> np.ndarray:
        """Pure fire decision for pass_num — NO state mutation.

        Computes exactly the mask step() would produce from the current
        state (same float ops), so the decision can be taken early (at the
        end of the previous optimizer step) to post the mask exchange
        off the critical path; step() later commits the state change.
        """
        if self.always_fire:
            return np.ones(self.sz, dtype=bool)
        norms = norms.astype(np.float32, copy=False)
        value_diff = np.abs(norms - self.last_sent_norm)
        thres_eff = (self.thres * self.horizon if self.adaptive
                     else np.full(self.sz, self.constant, dtype=np.float32))
        return (value_diff >= thres_eff) | (pass_num <
                                            self.initial_comm_passes)

    def step(self, norms: np.ndarray, pass_num: int) -> np.ndarray:
        """Evaluate the trigger for one pass; returns fire mask (bool[sz])."""
        norms = norms.astype(np.float32, copy=False)
        if self.always_fire:
            fire = np.ones(self.sz, dtype=bool)
        else:
            value_diff = np.abs(norms - self.last_sent_norm)
            iter_diff = np.float32(pass_num) - self.last_sent_iter
            if self.adaptive:
                self.thres = self.thres * self.horizon
            else:
                self.thres = np.full(self.sz, self.constant, dtype=np.float32)
            fire = (value_diff >= self.thres) | (pass_num <
                                                 self.initial_comm_passes)
        if fire.any():
            value_diff = np.abs(norms - self.last_sent_norm)
            iter_diff = np.maximum(
                np.float32(pass_num) - self.last_sent_iter, 1.0)
            new_slope = (value_diff / iter_diff).astype(np.float32)
            shifted = np.roll(self.slopes, -1, axis=1)
            shifted[:, -1] = new_slope
            self.slopes = np.where(fire[:, None], shifted, self.slopes)
            slope_avg = self.slopes.mean(axis=1, dtype=np.float32)
            if self.adaptive and not self.always_fire:
                self.thres = np.where(fire, slope_avg, self.thres)
            self.last_sent_norm = np.where(fire, norms, self.last_sent_norm)
            self.last_sent_iter = np.where(
                fire, np.float32(pass_num), self.last_sent_iter)
        self.num_events += 2 * int(fire.sum())
        self.last_norms = norms.copy()
        self.last_fired = fire.copy()
        return fire

    # -- checkpointing -----------------------------------------------------
    def state_dict(self) -> dict:
        return {
            "thres": self.thres.copy(),
            "last_sent_norm": self.last_sent_norm.copy(),
            "last_sent_iter": self.last_sent_iter.copy(),
            "slopes": self.slopes.copy(),
            "num_events": self.num_events,
        }

    def load_state_dict(self, d: dict) -> None:
        self.thres = d["thres"].astype(np.float32).copy()
        self.last_sent_norm = d["last_sent_norm"].astype(np.float32).copy()
        self.last_sent_iter = d["last_sent_iter"].astype(np.float32).copy()
        self.slopes = d["slopes"].astype(np.float32).copy()
        self.num_events = int(d["num_events"])


class GpuTriggerController:
    """Device-resident controller: state lives in GPU tensors and the whole
    trigger evaluation + threshold adaptation runs in the HIP kernel
    (csrc/engine.hip trigger_update), fused with the SGD-step kernel's
    norm output — only the fire mask (sz bytes) crosses to the host, to
    size the RCCL payloads. Kernel parity with TriggerController is
    covered by tests/test_gpu_numerics.py::test_trigger_update_matches_host_controller.
    """

    def __init__(self, sz: int, device, adaptive: bool, horizon: float,
                 constant: float, sent_history: int = 2,
                 initial_comm_passes: int = 30, always_fire: bool = False):
        import torch

        self.sz = sz
        self.device = device
        self.adaptive = adaptive
        self.horizon = float(horizon)
        self.constant = float(constant)
        self.sent_history = sent_history
        self.initial_comm_passes = initial_comm_passes
        self.always_fire = always_fire

        f32 = dict(dtype=torch.float32, device=device)
        self.thres = torch.zeros(sz, **f32)
        self.last_sent_norm = torch.zeros(sz, **f32)
        self.last_sent_iter = torch.zeros(sz, **f32)
        self.slopes = torch.zeros(sz * sent_history, **f32)
        self._num_events = torch.zeros(1, dtype=torch.int32, device=device)
        self.last_norms = None   # device sqnorms of the last step (for trace)
        self.last_fired = None

    @property
    def num_events(self) -> int:
        return int(self._num_events.item())

    @num_events.setter
    def num_events(self, v: int):
        self._num_events.fill_(int(v))

    def decide_device(self, norms_sq, pass_num: int):
        """Pure fire decision (no state mutation) — the device counterpart
        of TriggerController.decide. Returns np bool[sz] (one small D2H)."""
        import numpy as _np

        if self.always_fire:
            return _np.ones(self.sz, dtype=bool)
        from ..ops.backend import native

        mask = native().trigger_decide(
            norms_sq, self.thres, self.last_sent_norm, pass_num,
            self.adaptive, self.horizon, self.constant,
            self.initial_comm_passes)
        return mask.cpu().numpy().astype(bool)

    def step_device(self, norms_sq, pass_num: int, need_mask: bool = True):
        """norms_sq: device fp32[sz] (squared L2 norms). Returns np bool[sz]
        (or None when need_mask=False — commit-only, no D2H sync)."""
        from ..ops.backend import native

        mask = native().trigger_update(
            norms_sq, self.thres, self.last_sent_norm, self.last_sent_iter,
            self.slopes, self._num_events, pass_num, self.adaptive,
            self.horizon, self.constant, self.initial_comm_passes,
            self.always_fire)
        self.last_norms = norms_sq
        if not need_mask:
            return None
        fired = mask.cpu().numpy().astype(bool)
        self.last_fired = fired
        return fired

    def trace_values(self):
        """(norms, thres) as numpy — only used when tracing is enabled."""
        import numpy as _np
        norms = _np.sqrt(self.last_norms.cpu().numpy()) \
            if self.last_norms is not None else _np.zeros(self.sz, _np.float32)
        return norms, self.thres.cpu().numpy()

    # -- checkpointing (same dict format as the host controller) ----------
    def state_dict(self) -> dict:
        return {
            "thres": self.thres.cpu().numpy(),
            "last_sent_norm": self.last_sent_norm.cpu().numpy(),
            "last_sent_iter": self.last_sent_iter.cpu().numpy(),
            "slopes": self.slopes.cpu().numpy().reshape(
                self.sz, self.sent_history),
            "num_events": self.num_events,
        }

    def load_state_dict(self, d: dict) -> None:
        import torch

        for name in ("thres", "last_sent_norm", "last_sent_iter"):
            getattr(self, name).copy_(
                torch.from_numpy(np.asarray(d[name], dtype=np.float32)))
        self.slopes.copy_(torch.from_numpy(
            np.asarray(d["slopes"], dtype=np.float32).reshape(-1)))
        self.num_events = int(d["num_events"])
