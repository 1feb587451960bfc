"""Checkpoint / resume (new capability; the reference has none — SURVEY.md §5).

Persists, per rank: flat params, SGD momentum, pass_num/epoch, the trigger
controller state (thres / last_sent_norm / last_sent_iter / slopes /
num_events), the stale inboxes (or dense neighbor replicas for spevent), the
spevent prev (last-sent-values) buffer, and the torch RNG states — exactly
the state the modes' semantics need to resume bit-identically.
"""

from __future__ import annotations

import dataclasses
import os

import torch


def save_checkpoint(path: str, cfg, epoch: int, pass_num: int, space, engine,
                    extra: dict | None = None) -> None:
    state = {
        "config": dataclasses.asdict(cfg),
        "epoch": epoch,
        "pass_num": pass_num,
        "param": space.param.detach().cpu(),
        "momentum": space.momentum.detach().cpu(),
        "names": space.names,
        "rng_cpu": torch.get_rng_state(),
        "engine": engine_state_dict(engine),
        "extra": extra or {},
    }
    if torch.cuda.is_available() and space.param.is_cuda:
        state["rng_cuda"] = torch.cuda.get_rng_state()
    tmp = path + ".tmp"
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    torch.save(state, tmp)
    os.replace(tmp, path)


def engine_state_dict(engine) -> dict:
    d: dict = {}
    ctrl = getattr(engine, "ctrl", None)
    if ctrl is not None:
        d["controller"] = ctrl.state_dict()
    for name in ("inbox_left", "inbox_right", "prev"):
        t = getattr(engine, name, None)
        if isinstance(t, torch.Tensor):
            d[name] = t.detach().cpu()
    d["num_events"] = engine.num_events
    return d


def load_checkpoint(path: str, space, engine, device) -> dict:
    state = torch.load(path, map_location="cpu", weights_only=False)
    space.param.copy_(state["param"].to(device))
    space.momentum.copy_(state["momentum"].to(device))
    ed = state.get("engine", {})
    ctrl = getattr(engine, "ctrl", None)
    if ctrl is not None and "controller" in ed:
        ctrl.load_state_dict(ed["controller"])
    for name in ("inbox_left", "inbox_right", "prev"):
        t = getattr(engine, name, None)
        if isinstance(t, torch.Tensor) and name in ed:
            t.copy_(ed[name].to(device))
    engine.num_events = ed.get("num_events", 0)
    torch.set_rng_state(state["rng_cpu"])
    if "rng_cuda" in state and torch.cuda.is_available():
        torch.cuda.set_rng_state(state["rng_cuda"])
    return state
