#!/usr/bin/env python3
"""Flagship benchmark: CIFAR-10 quirk-ResNet EventGraD training step.

Measures whole-job samples/sec for the BASELINE.json headline config
("CIFAR-10 ResNet EventGraD (dcifar10/event) ring"): the reference's coded
model ResNet<BasicBlock>({2,2,2,2}) with its make_layer quirk — 17,444,682
params / 86 tensors — SGD lr 1e-2 momentum 0.9, adaptive
event trigger (horizon 1.01, 30 warmup passes), synthetic CIFAR-shaped data
(32x32x3, random-init weights; no network in this environment). Per-GPU
batch is fixed at 256 (the reference's global batch at 1 rank), so scaling
is WEAK; the driver computes scaling efficiency from per-N runs.

Contract: one rank per GPU via torch.distributed.run; rank 0 prints ONE
JSON line. Timed region = K full training steps (data->device, trigger +
ring gossip, forward, loss, backward, neighbor averaging, fused SGD step)
bracketed by barrier + torch.cuda.synchronize on both sides; value uses the
MAX elapsed over ranks.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch

from eventgrad_amd.config import preset
from eventgrad_amd.models import build_model
from eventgrad_amd.ops import functional as O
from eventgrad_amd.parallel import FlatParamSpace, build_engine, init_distributed


def preflight(rank: int, world: int, device) -> int:
    """5-step RCCL smoke of all four modes with per-rank digests.

    VERDICT r1 item 1(b): makes a scaling-run failure diagnosable in one
    shot — every rank reports param/mask/event digests per mode, rank 0
    prints one JSON line per mode, and any non-finite value fails the run.
    """
    import torch.distributed as dist

    ok = True
    for mode in ("cent", "decent", "event", "spevent"):
        cfg = preset("dcifar10-event")
        cfg.mode = mode
        cfg.data.global_batch = None
        cfg.data.batch_size = 32
        cfg.trigger.initial_comm_passes = 2   # passes 3-5 use dynamic masks
        torch.manual_seed(cfg.seed)
        model = build_model(cfg.model).to(device)
        model.train()
        space = FlatParamSpace(model, device)
        engine = build_engine(space, cfg, rank, world, device)
        g = torch.Generator(device="cpu").manual_seed(1234 + rank)
        x = torch.randn(32, 3, 32, 32, generator=g).to(device)
        y = torch.randint(0, 10, (32,), generator=g).to(device)
        fired_hist = []
        for p in range(1, 6):
            engine.begin_pass(p)
            space.zero_grad()
            logits = model(x)
            loss = O.nll_of_logits(logits, y)
            loss.backward()
            engine.after_backward()
            engine.step()
            fired = getattr(engine, "_fired_l", None)
            fired_hist.append(-1 if fired is None else len(fired))
        # modes share one process group and fixed tags: the trailing
        # lookahead mask exchange must drain before the next mode posts
        engine.drain()
        torch.cuda.synchronize()
        if world > 1:
            dist.barrier()
        digest = {
            "rank": rank,
            "param_sqnorm": round(float(space.param.square().sum()), 4),
            "param_head": [round(float(v), 6)
                           for v in space.param[:4].tolist()],
            "loss": round(float(loss.detach()), 5),
            "num_events": engine.num_events,
            "recv_left_fired_per_pass": fired_hist,
            "finite": bool(torch.isfinite(space.param).all()),
        }
        gathered = [None] * world
        if world > 1:
            dist.all_gather_object(gathered, digest)
        else:
            gathered = [digest]
        mode_ok = all(d["finite"] for d in gathered)
        ok = ok and mode_ok
        if rank == 0:
            print(json.dumps({"preflight": mode,
                              "backend": dist.get_backend()
                              if dist.is_initialized() else "none",
                              "world": world, "ok": mode_ok,
                              "ranks": gathered}), flush=True)
    if world > 1:
        dist.barrier()
        torch.distributed.destroy_process_group()
    return 0 if ok else 1


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--batch", type=int, default=256, help="per-GPU batch")
    ap.add_argument("--mode", default="event",
                    choices=["event", "decent", "cent", "spevent", "serial"])
    ap.add_argument("--model", default="resnet18q")
    ap.add_argument("--no-graph", action="store_true",
                    help="disable hipGraph capture of fwd+bwd")
    ap.add_argument("--backend", default=None,
                    help="dist backend override (testing: gloo lets two "
                         "ranks share one GPU; default nccl=RCCL)")
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"],
                    help="GPU compute path (fp32 = torch/MIOpen "
                         "full-precision; used for parity runs)")
    ap.add_argument("--preflight", action="store_true",
                    help="multi-GPU RCCL smoke: 5 steps of every mode with "
                         "per-rank digests printed (run under torchrun on "
                         "any multi-GPU box before a scaling run)")
    args = ap.parse_args()

    cfg = preset("dcifar10-event")
    cfg.mode = args.mode
    cfg.model = args.model
    cfg.data.global_batch = None
    cfg.data.batch_size = args.batch
    cfg.compute_dtype = args.dtype
    O.set_compute_dtype(args.dtype)

    rank, world, device = init_distributed("auto", backend=args.backend)
    if device.type != "cuda":
        raise SystemExit("bench.py requires a GPU")
    if args.preflight:
        return preflight(rank, world, device)
    torch.manual_seed(cfg.seed)

    model = build_model(cfg.model).to(device)
    model.train()
    space = FlatParamSpace(model, device)
    engine = build_engine(space, cfg, rank, world, device)

    # synthetic data, resident batches (data=synthetic per BASELINE contract)
    g = torch.Generator(device="cpu").manual_seed(1234 + rank)
    n_resident = 8
    shape = ((1, 28, 28) if args.model in ("cnn1", "cnn2", "mlp")
             else (3, 32, 32))
    xs = [torch.randn(args.batch, *shape, generator=g).to(device)
          for _ in range(n_resident)]
    ys = [torch.randint(0, 10, (args.batch,), generator=g).to(device)
          for _ in range(n_resident)]

    pass_num = 0
    graph = None
    if not args.no_graph and args.dtype == "bf16":
        from eventgrad_amd.train.graphstep import FwdBwdGraph, can_graph
        if can_graph(model, device):
            graph = FwdBwdGraph(model, space, tuple(xs[0].shape), device)
            # capture BEFORE any RCCL p2p is in flight (graph capture with
            # outstanding comm on other streams is the risky combination)
            graph.step(xs[0], ys[0])
            torch.cuda.synchronize()

    def step():
        nonlocal pass_num
        pass_num += 1
        x, y = xs[pass_num % n_resident], ys[pass_num % n_resident]
        engine.begin_pass(pass_num)
        if graph is not None:
            logits, loss = graph.step(x, y)
        else:
            space.zero_grad()
            logits = model(x)
            loss = O.nll_of_logits(logits, y)
            loss.backward()
        engine.after_backward()
        engine.step()
        return loss

    def barrier_sync():
        if world > 1:
            torch.distributed.barrier()
        torch.cuda.synchronize()

    for _ in range(args.warmup):
        step()
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = step()
    barrier_sync()
    elapsed = time.perf_counter() - t0
    engine.drain()  # complete the last step's lookahead mask exchange

    if world > 1:
        t = torch.tensor([elapsed], device=device)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        samples = float(world * args.batch * args.steps)
        out = {
            "metric": "samples_per_sec",
            "value": round(samples / elapsed, 2),
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(1000.0 * elapsed / args.steps, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": ("resnet18-quirk (ref ResNet<BasicBlock>{2,2,2,2}, "
                          "17444682 params / 86 tensors)"
                          if args.model == "resnet18q" else args.model),
                "mode": args.mode,
                "global_batch": world * args.batch,
                "per_gpu_batch": args.batch,
                "seq_len": None,
                "parallelism": f"eventgrad-ring dp{world}",
                "optimizer": "sgd lr=1e-2 momentum=0.9",
                "trigger": "adaptive horizon=1.01 warmup=30",
                "hip_graph": graph is not None,
                "final_loss": round(float(loss.item()), 4),
            },
        }
        print(json.dumps(out), flush=True)
    if world > 1:
        torch.distributed.destroy_process_group()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
