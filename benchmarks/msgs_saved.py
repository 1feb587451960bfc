#!/usr/bin/env python3
"""Headline metric harness: messages saved at matched accuracy.

Runs the always-communicate ring (decent — the baseline that sends
2*num_tensors messages per rank per pass) and EventGraD (event, optionally
spevent) on the SAME data/model/seed, and reports the saved-message
percentage together with both runs' accuracies — the reference's headline
experiment (README.md:4: ~70% saved on MNIST / ~60% on CIFAR-10 at matched
accuracy; counting rule in BASELINE.md).

Self-spawning (gloo on CPU; one process per GPU with RCCL when run under
torchrun with CUDA). Examples:

  python benchmarks/msgs_saved.py --experiment mnist --world 4
  python benchmarks/msgs_saved.py --experiment cifar10 --world 4 --epochs 10
  torchrun --standalone --nproc-per-node 8 benchmarks/msgs_saved.py \
      --experiment cifar10 --launched
"""

from __future__ import annotations

import argparse
import json
import os
import sys

import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def build_cfg(experiment: str, mode: str, args):
    from eventgrad_amd.config import preset

    if experiment == "mnist":
        cfg = preset("dmnist-event")           # CNN-2, batch 64, lr .05
        cfg.data.dataset = "synthetic-mnist"
        cfg.data.synthetic_train_samples = args.train_samples or 8192
        cfg.epochs = args.epochs or 10
    else:
        cfg = preset("dcifar10-event")         # quirk ResNet, global 256
        cfg.data.dataset = "synthetic"
        cfg.data.synthetic_train_samples = args.train_samples or 8192
        cfg.data.augment = False               # synthetic prototypes
        cfg.epochs = args.epochs or 10
    cfg.data.synthetic_test_samples = 1024
    cfg.data.synthetic_noise = args.noise
    cfg.data.label_noise = args.label_noise
    cfg.seed = args.seed
    if args.global_batch:
        cfg.data.global_batch = args.global_batch
        cfg.data.batch_size = args.global_batch
    cfg.mode = mode
    cfg.trigger.adaptive = not args.constant
    cfg.trigger.horizon = args.horizon
    cfg.trigger.constant = args.thres
    cfg.trigger.initial_comm_passes = args.warmup_passes
    cfg.topk_percent = args.topk_percent
    cfg.device = args.device
    if args.backend:
        cfg.dist_backend = args.backend
    return cfg


def run_one(mode, args, outfile):
    from eventgrad_amd.train.trainer import Trainer

    cfg = build_cfg(args.experiment, mode, args)
    tr = Trainer(cfg)
    m = tr.train()
    if tr.rank == 0:
        with open(outfile, "w") as f:
            json.dump(m.summary(), f)
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()


def worker(rank, world, port, args_d, outdir):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1")
    args = argparse.Namespace(**args_d)
    for mode in args.modes:
        os.environ["MASTER_PORT"] = str(port)
        port += 1
        run_one(mode, args, os.path.join(outdir, f"{mode}.json"))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--experiment", choices=["mnist", "cifar10"],
                    default="mnist")
    ap.add_argument("--world", type=int, default=4)
    ap.add_argument("--epochs", type=int, default=None)
    ap.add_argument("--train-samples", type=int, default=None)
    ap.add_argument("--noise", type=float, default=0.8)
    ap.add_argument("--label-noise", type=float, default=0.0,
                    help="fraction of train labels flipped (test clean); "
                         "holds accuracy in a non-saturating band")
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--global-batch", type=int, default=None,
                    help="override the global batch (more passes per epoch)")
    ap.add_argument("--horizon", type=float, default=1.01)
    ap.add_argument("--constant", action="store_true",
                    help="use static threshold instead of adaptive")
    ap.add_argument("--thres", type=float, default=5e-4)
    ap.add_argument("--warmup-passes", type=int, default=30)
    ap.add_argument("--topk-percent", type=float, default=1.0)
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--backend", default=None,
                    help="force dist backend (e.g. gloo for N ranks on one "
                         "GPU; the wire stages through host)")
    ap.add_argument("--modes", nargs="+",
                    default=["decent", "event"])
    ap.add_argument("--outdir", default="benchmarks/out")
    ap.add_argument("--launched", action="store_true",
                    help="already under torchrun; do not self-spawn")
    ap.add_argument("--port", type=int, default=29650,
                    help="rendezvous base port (change for concurrent runs)")
    args = ap.parse_args()
    os.makedirs(args.outdir, exist_ok=True)

    if args.launched:
        for mode in args.modes:
            run_one(mode, args, os.path.join(args.outdir, f"{mode}.json"))
    else:
        mp.start_processes(worker,
                           args=(args.world, args.port, vars(args), args.outdir),
                           nprocs=args.world, start_method="spawn", join=True)

    results = {}
    for mode in args.modes:
        with open(os.path.join(args.outdir, f"{mode}.json")) as f:
            results[mode] = json.load(f)
    base = results.get("decent")
    report = {"experiment": args.experiment, "world": args.world,
              "trigger": ("adaptive h=" + str(args.horizon)
                          if not args.constant else f"const {args.thres}")}
    for mode, r in results.items():
        report[mode] = {
            "messages_saved_pct": r["messages_saved_pct"],
            "events": r["num_events_total"],
            "possible": r["messages_possible"],
            "final_train_acc": r["epoch_train_acc"][-1],
            "test_accuracy": r["test_accuracy"],
        }
        if base and mode != "decent" and base["test_accuracy"]:
            report[mode]["test_acc_delta_vs_decent"] = round(
                r["test_accuracy"] - base["test_accuracy"], 3)
    print(json.dumps(report, indent=2))
    with open(os.path.join(args.outdir,
                           f"report_{args.experiment}.json"), "w") as f:
        json.dump(report, f, indent=2)


if __name__ == "__main__":
    main()
