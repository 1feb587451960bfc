"""CLI for all training modes.

Flag form (every reference hard-coded constant exposed):

  torchrun --standalone --nproc-per-node 8 -m eventgrad_amd.train \
      --mode event --preset dcifar10-event --thres-adaptive --horizon 1.01

Reference-compatible positional form lives in eventgrad_amd.apps.* (e.g.
``python -m eventgrad_amd.apps.dmnist_event <file_write> <thres_type>
<horizon|constant>`` mirroring dmnist/event/README.md:30-57).
"""

from __future__ import annotations

import argparse
import json

from ..config import PRESETS, RunConfig, preset


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(prog="eventgrad_amd.train")
    p.add_argument("--preset", choices=PRESETS, default=None)
    p.add_argument("--mode", choices=["cent", "decent", "event", "spevent",
                                      "serial"], default=None)
    p.add_argument("--model", default=None)
    p.add_argument("--dataset", default=None,
                   help="mnist | cifar10 | synthetic | synthetic-mnist")
    p.add_argument("--data-path", default=None)
    p.add_argument("--epochs", type=int, default=None)
    p.add_argument("--batch-size", type=int, default=None)
    p.add_argument("--global-batch", type=int, default=None)
    p.add_argument("--synthetic-samples", type=int, default=None,
                   help="synthetic dataset train size (no-network runs)")
    p.add_argument("--lr", type=float, default=None)
    p.add_argument("--momentum", type=float, default=None)
    p.add_argument("--seed", type=int, default=None)
    # trigger
    p.add_argument("--thres-adaptive", action="store_true", default=None)
    p.add_argument("--thres-constant", type=float, default=None,
                   help="static threshold (implies non-adaptive)")
    p.add_argument("--horizon", type=float, default=None)
    p.add_argument("--warmup-passes", type=int, default=None)
    p.add_argument("--sent-history", type=int, default=None)
    p.add_argument("--topk-percent", type=float, default=None)
    # io
    p.add_argument("--trace", action="store_true", default=None,
                   help="write send{rank}.txt/recv{rank}.txt (ref file_write)")
    p.add_argument("--trace-dir", default=None)
    p.add_argument("--checkpoint", dest="checkpoint_path", default=None)
    p.add_argument("--checkpoint-every", type=int, default=None)
    p.add_argument("--resume", action="store_true", default=None)
    p.add_argument("--device", default=None)
    p.add_argument("--compute-dtype", choices=["bf16", "fp32"], default=None,
                   help="GPU compute path: bf16 = native HIP kernels "
                        "(default), fp32 = full-precision torch/MIOpen")
    p.add_argument("--no-eval", action="store_true", default=None)
    p.add_argument("--json-out", default=None,
                   help="write metrics summary JSON here (rank 0)")
    return p


def config_from_args(args) -> RunConfig:
    cfg = preset(args.preset) if args.preset else RunConfig()
    for name, attr in [("mode", "mode"), ("model", "model"),
                       ("epochs", "epochs"), ("seed", "seed"),
                       ("device", "device"),
                       ("checkpoint_path", "checkpoint_path"),
                       ("trace_dir", "trace_dir")]:
        v = getattr(args, name)
        if v is not None:
            setattr(cfg, attr, v)
    if args.topk_percent is not None:
        cfg.topk_percent = args.topk_percent
    if args.checkpoint_every is not None:
        cfg.checkpoint_every_epochs = args.checkpoint_every
    if args.resume:
        cfg.resume = True
    if args.trace:
        cfg.trace = True
    if args.no_eval:
        cfg.eval_at_end = False
    if args.compute_dtype is not None:
        cfg.compute_dtype = args.compute_dtype
    if args.dataset is not None:
        cfg.data.dataset = args.dataset
    if args.data_path is not None:
        cfg.data.data_path = args.data_path
    if args.batch_size is not None:
        cfg.data.batch_size = args.batch_size
    if args.global_batch is not None:
        cfg.data.global_batch = args.global_batch
    if args.synthetic_samples is not None:
        cfg.data.synthetic_train_samples = args.synthetic_samples
        cfg.data.synthetic_test_samples = max(64, args.synthetic_samples // 8)
    if args.lr is not None:
        cfg.optim.lr = args.lr
    if args.momentum is not None:
        cfg.optim.momentum = args.momentum
    if args.thres_constant is not None:
        cfg.trigger.adaptive = False
        cfg.trigger.constant = args.thres_constant
    if args.thres_adaptive:
        cfg.trigger.adaptive = True
    if args.horizon is not None:
        cfg.trigger.horizon = args.horizon
    if args.warmup_passes is not None:
        cfg.trigger.initial_comm_passes = args.warmup_passes
    if args.sent_history is not None:
        cfg.trigger.sent_history = args.sent_history
    return cfg


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    cfg = config_from_args(args)
    from .trainer import Trainer

    tr = Trainer(cfg)
    metrics = tr.train()
    if tr.rank == 0:
        summary = metrics.summary()
        print(json.dumps(summary), flush=True)
        if args.json_out:
            with open(args.json_out, "w") as f:
                json.dump(summary, f, indent=2)
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
