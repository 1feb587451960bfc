#!/usr/bin/env python3
"""bf16-vs-fp32 convergence-parity artifact.

The reference trains fp32 on CPU (dcifar10/event/event.cpp:39 kCPU). The
MI355X-native hot path computes in bf16 (NHWC MFMA kernels) with fp32
parameters/wire. This script trains the flagship model end-to-end twice on
one GPU with the SAME seed/config — compute_dtype "bf16" (native HIP
kernels) and "fp32" (full-precision torch/MIOpen path, RunConfig's other
compute_dtype value) — and records the loss/accuracy trajectories, so the
bf16 headline throughput number carries a precision-parity citation.

Run on a GPU box:  python benchmarks/precision_parity.py [--epochs 8]
Writes benchmarks/precision_parity.json.
"""

from __future__ import annotations

import argparse
import json
import os
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, os.path.dirname(HERE))


def run(dtype: str, args) -> dict:
    import torch

    from eventgrad_amd.config import DataConfig, OptimConfig, RunConfig
    from eventgrad_amd.train.trainer import Trainer

    cfg = RunConfig(
        mode="serial", model="resnet18q", epochs=args.epochs, device="cuda",
        compute_dtype=dtype,
        data=DataConfig(dataset="synthetic", batch_size=256,
                        synthetic_train_samples=args.train_samples,
                        synthetic_test_samples=1024,
                        synthetic_noise=args.noise, label_noise=0.1),
        optim=OptimConfig(lr=1e-2, momentum=0.9))
    tr = Trainer(cfg)
    m = tr.train()
    del tr
    torch.cuda.empty_cache()
    return {
        "dtype": dtype,
        "epoch_train_acc": m.epoch_train_acc,
        "final_train_loss": round(m.final_train_loss, 4),
        "test_accuracy": m.test_accuracy,
        "test_loss": round(m.test_loss, 4) if m.test_loss else None,
        "train_time_s": round(m.train_time_s, 2),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--epochs", type=int, default=8)
    ap.add_argument("--train-samples", type=int, default=4096)
    ap.add_argument("--noise", type=float, default=1.2)
    args = ap.parse_args()

    out = {"config": vars(args),
           "runs": [run("fp32", args), run("bf16", args)]}
    fp32, bf16 = out["runs"]
    out["test_acc_delta_bf16_minus_fp32"] = round(
        bf16["test_accuracy"] - fp32["test_accuracy"], 3)
    out["final_acc_curves_max_gap"] = round(max(
        abs(a - b) for a, b in zip(bf16["epoch_train_acc"],
                                   fp32["epoch_train_acc"])), 3)
    path = os.path.join(HERE, "precision_parity.json")
    with open(path, "w") as f:
        json.dump(out, f, indent=2)
    print(json.dumps(out, indent=2))


if __name__ == "__main__":
    main()
