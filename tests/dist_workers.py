"""Importable worker functions for spawned multi-process tests."""

from __future__ import annotations

import json
import os

import torch

from util_dist import init_env


def _small_cfg(mode: str, **kw):
    from eventgrad_amd.config import (DataConfig, OptimConfig, RunConfig,
                                      TriggerConfig)
    trig = kw.pop("trigger", TriggerConfig(adaptive=True, horizon=1.01,
                                           initial_comm_passes=5))
    cfg = RunConfig(
        mode=mode, model=kw.pop("model", "cnn2"),
        epochs=kw.pop("epochs", 2), device="cpu", trigger=trig,
        data=DataConfig(dataset=kw.pop("dataset", "synthetic-mnist"),
                        batch_size=kw.pop("batch_size", 32),
                        synthetic_train_samples=kw.pop("n_train", 256),
                        synthetic_test_samples=64),
        optim=OptimConfig(lr=kw.pop("lr", 0.05),
                          momentum=kw.pop("momentum", 0.0)),
        eval_at_end=False,
        **kw,
    )
    return cfg


def train_mode_worker(rank, world, port, mode, outdir, kw_json="{}"):
    """Train a small run; dump final flat params + metrics per rank."""
    init_env(rank, world, port)
    kw = json.loads(kw_json)
    cfg = _small_cfg(mode, **kw)
    from eventgrad_amd.train.trainer import Trainer

    tr = Trainer(cfg)
    m = tr.train()
    torch.save(
        {"param": tr.space.param.clone(), "metrics": m.summary()},
        os.path.join(outdir, f"{mode}_r{rank}.pt"),
    )
    torch.distributed.destroy_process_group()


def event_equals_decent_worker(rank, world, port, outdir):
    """Run decent, then event with constant threshold 0 + no warmup; both
    always fire, so trajectories must be IDENTICAL (dmnist/event/README.md:59-60)."""
    init_env(rank, world, port)
    from eventgrad_amd.config import TriggerConfig
    from eventgrad_amd.train.trainer import Trainer

    params = {}
    for cycle, (mode, trig) in enumerate([
        ("decent", TriggerConfig()),
        ("event", TriggerConfig(adaptive=False, constant=0.0,
                                initial_comm_passes=0)),
    ]):
        # fresh port per init cycle: re-binding the TCPStore on the same
        # port right after destroy_process_group races (observed hangs)
        os.environ["MASTER_PORT"] = str(port + 1 + cycle)
        cfg = _small_cfg(mode, trigger=trig, epochs=2)
        tr = Trainer(cfg)
        tr.train()
        params[mode] = tr.space.param.clone()
        torch.distributed.destroy_process_group()
    same = torch.equal(params["decent"], params["event"])
    torch.save({"identical": same},
               os.path.join(outdir, f"eqdec_r{rank}.pt"))


def event_twice_deterministic_worker(rank, world, port, outdir):
    """Run the same post-warmup sparse-mask event config twice; the matched
    mask+payload protocol must be fully deterministic (identical params and
    identical event counts), at any world size."""
    init_env(rank, world, port)
    from eventgrad_amd.train.trainer import Trainer

    results = []
    for cycle in range(2):
        os.environ["MASTER_PORT"] = str(port + cycle)
        cfg = _small_cfg("event", epochs=2, momentum=0.9)
        tr = Trainer(cfg)
        m = tr.train()
        results.append((tr.space.param.clone(), m.num_events_total))
        torch.distributed.destroy_process_group()
    torch.save({"identical": torch.equal(results[0][0], results[1][0]),
                "events0": results[0][1], "events1": results[1][1]},
               os.path.join(outdir, f"det_r{rank}.pt"))


def cent_equals_fullbatch_worker(rank, world, port, outdir):
    """cent (allreduce-averaged grads over equal shards) must equal a serial
    run on the concatenated data with the same lr (cent.cpp:130-145
    semantics), because mean-of-shard-grads == full-batch grad."""
    init_env(rank, world, port)
    from eventgrad_amd.config import (DataConfig, OptimConfig, RunConfig)
    from eventgrad_amd.train.trainer import Trainer

    cfg = RunConfig(
        mode="cent", model="mlp", epochs=3, device="cpu",
        data=DataConfig(dataset="synthetic-mnist", batch_size=0, shuffle=False,
                        synthetic_train_samples=128,
                        synthetic_test_samples=32),
        optim=OptimConfig(lr=0.01), eval_at_end=False)
    tr = Trainer(cfg)
    tr.train()
    torch.save({"param": tr.space.param.clone()},
               os.path.join(outdir, f"cent_r{rank}.pt"))
    torch.distributed.destroy_process_group()


def checkpoint_resume_worker(rank, world, port, outdir):
    """event mode: 1 epoch + checkpoint + 1 epoch  ==  2 epochs straight."""
    init_env(rank, world, port)
    from eventgrad_amd.train.trainer import Trainer

    ck = os.path.join(outdir, f"ck_r{rank}.pt")

    cfg = _small_cfg("event", epochs=2, momentum=0.9)
    tr = Trainer(cfg)
    tr.train()
    ref = tr.space.param.clone()
    torch.distributed.destroy_process_group()

    # intermediate run must not run the closing consensus allreduce, so the
    # checkpoint holds exactly the straight run's end-of-epoch-1 state.
    # Fresh port per init cycle (TCPStore rebind race -> hangs).
    os.environ["MASTER_PORT"] = str(port + 1)
    cfg1 = _small_cfg("event", epochs=1, momentum=0.9)
    cfg1.checkpoint_path = ck
    cfg1.final_consensus = False
    tr1 = Trainer(cfg1)
    tr1.train()
    torch.distributed.destroy_process_group()

    os.environ["MASTER_PORT"] = str(port + 2)
    cfg2 = _small_cfg("event", epochs=2, momentum=0.9)
    cfg2.checkpoint_path = ck
    cfg2.resume = True
    tr2 = Trainer(cfg2)
    tr2.train()
    torch.distributed.destroy_process_group()

    torch.save({"ref": ref, "resumed": tr2.space.param.clone()},
               os.path.join(outdir, f"ckres_r{rank}.pt"))


def checkpoint_resume_bn_worker(rank, world, port, outdir):
    """Resume identity for a BatchNorm model (resnet20): params AND the BN
    running stats (registered buffers, outside FlatParamSpace) must round-trip
    through the checkpoint, so post-resume eval matches the straight run."""
    init_env(rank, world, port)
    from eventgrad_amd.train.trainer import Trainer

    kw = dict(model="resnet20", dataset="synthetic", batch_size=16,
              n_train=64, lr=0.05, momentum=0.9)
    ck = os.path.join(outdir, "ckbn.pt")  # single path; Trainer adds .rank{r}

    def _bufs(model):
        # sync python-side mirrors (num_batches_tracked) before harvesting
        for m in model.modules():
            if hasattr(m, "sync_buffers_for_save"):
                m.sync_buffers_for_save()
        return {k: v.detach().clone() for k, v in model.named_buffers()}

    cfg = _small_cfg("event", epochs=2, **kw)
    tr = Trainer(cfg)
    tr.train()
    ref_param = tr.space.param.clone()
    ref_bufs = _bufs(tr.model)
    ref_eval = tr.evaluate() if rank == 0 else None
    ref_loss = tr.metrics.test_loss if rank == 0 else None
    torch.distributed.destroy_process_group()

    os.environ["MASTER_PORT"] = str(port + 1)
    cfg1 = _small_cfg("event", epochs=1, **kw)
    cfg1.checkpoint_path = ck
    cfg1.final_consensus = False
    tr1 = Trainer(cfg1)
    tr1.train()
    torch.distributed.destroy_process_group()

    os.environ["MASTER_PORT"] = str(port + 2)
    cfg2 = _small_cfg("event", epochs=2, **kw)
    cfg2.checkpoint_path = ck
    cfg2.resume = True
    tr2 = Trainer(cfg2)
    tr2.train()
    res_bufs = _bufs(tr2.model)
    bufs_equal = all(torch.equal(ref_bufs[k], res_bufs[k])
                     for k in ref_bufs)
    res_eval = tr2.evaluate() if rank == 0 else None
    res_loss = tr2.metrics.test_loss if rank == 0 else None
    torch.distributed.destroy_process_group()

    torch.save({"ref": ref_param, "resumed": tr2.space.param.clone(),
                "bufs_equal": bufs_equal, "n_bufs": len(ref_bufs),
                "ref_eval": ref_eval, "res_eval": res_eval,
                "ref_loss": ref_loss, "res_loss": res_loss},
               os.path.join(outdir, f"ckbn_r{rank}.pt"))
