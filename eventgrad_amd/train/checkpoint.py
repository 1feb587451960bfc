"""Checkpoint / resume (new capability; the reference has none — SURVEY.md §5).

Persists, per rank: flat params, SGD momentum, model buffers (BatchNorm
running_mean/running_var — registered buffers are NOT named_parameters and
so live outside FlatParamSpace, resnet.hpp:18-19), pass_num/epoch, the
trigger controller state (thres / last_sent_norm / last_sent_iter / slopes /
num_events), the stale inboxes (or dense neighbor replicas for spevent), the
spevent prev (last-sent-values) buffer, and the torch RNG states — exactly
the state the modes' semantics need to resume bit-identically.

Multi-rank runs write one file per rank: ``rank_checkpoint_path`` appends a
``.rank{r}`` suffix when world > 1, so a single ``--checkpoint`` argument is
safe under torchrun (each rank's pre-consensus params / trigger state /
inboxes differ and must not clobber each other).
"""

from __future__ import annotations

import dataclasses
import os

import torch


def rank_checkpoint_path(path: str, rank: int, world: int) -> str:
    """Per-rank checkpoint file name; unchanged for single-rank runs."""
    if world <= 1:
        return path
    return f"{path}.rank{rank}"


def save_checkpoint(path: str, cfg, epoch: int, pass_num: int, space, engine,
                    extra: dict | None = None, model=None) -> None:
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
    if model is not None:
        # non-parameter state: BN running stats + num_batches_tracked
        for m in model.modules():
            if hasattr(m, "sync_buffers_for_save"):
                m.sync_buffers_for_save()
        state["buffers"] = {name: buf.detach().cpu().clone()
                            for name, buf in model.named_buffers()}
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


def load_checkpoint(path: str, space, engine, device, model=None) -> dict:
    state = torch.load(path, map_location="cpu", weights_only=False)
    space.param.copy_(state["param"].to(device))
    space.momentum.copy_(state["momentum"].to(device))
    if model is not None and "buffers" in state:
        bufs = dict(model.named_buffers())
        for name, saved in state["buffers"].items():
            if name in bufs:
                bufs[name].copy_(saved.to(bufs[name].device))
        for m in model.modules():
            if hasattr(m, "sync_buffers_after_load"):
                m.sync_buffers_after_load()
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
