"""Configuration system.

The reference hardcodes nearly everything (per-binary constants, positional
argv); this module exposes every one of those knobs as a typed config field
with the reference value as the default (SURVEY.md §5 "Config / flag system").

Reference constant provenance:
  - sent_history=2            dmnist/event/event.cpp:103
  - initial_comm_passes=30    dmnist/event/event.cpp:262
  - thres_type 0/1, horizon/constant   dmnist/event/event.cpp:89-100
  - topk_percent              dcifar10/spevent/spevent.cpp:60
  - MNIST event: batch 64, 10 epochs, SGD lr 0.05   event.cpp:145,227-230,255
  - MNIST cent: full-shard batch, 250 epochs, lr 1e-2   cent.cpp:62-75,95
  - MNIST decent: 50 epochs, lr 1e-2   decent.cpp:81-139
  - CIFAR-10: global batch 256, 20 epochs, SGD lr 1e-2 momentum 0.9
    dcifar10/event/event.cpp:29-42,196-200
"""

from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class TriggerConfig:
    """Event-trigger + adaptive-threshold controller parameters.

    Semantics (dmnist/event/event.cpp:324-392):
      each pass, per tensor i:
        value_diff = |  ||theta_i||_2  - last_sent_norm[i] |
        pre-update: thres[i] *= horizon   (adaptive)   or thres[i] = constant
        fire iff value_diff >= thres[i] or pass_num < initial_comm_passes
        on fire: slope hist shift; new slope = value_diff/iter_diff;
                 adaptive: thres[i] = mean(slopes); update last_sent_*.
    """

    adaptive: bool = True          # ref thres_type: 1 adaptive, 0 constant
    horizon: float = 1.01          # adaptive threshold growth per pass
    constant: float = 5e-4         # static threshold (ref README ~5e-4)
    sent_history: int = 2          # slope history length
    initial_comm_passes: int = 30  # warmup passes that always fire


@dataclass
class OptimConfig:
    """Plain/momentum SGD (the only optimizer the reference uses)."""

    lr: float = 0.01
    momentum: float = 0.0
    weight_decay: float = 0.0
    # divide allreduced gradients by world size in cent mode (cent.cpp:140)
    average_grads: bool = True


@dataclass
class DataConfig:
    dataset: str = "cifar10"      # mnist | cifar10 | synthetic
    data_path: Optional[str] = None  # None => synthetic data (no-network env)
    # per-rank batch size. The reference CIFAR binaries use global 256 split
    # over ranks (dcifar10/event/event.cpp:91); MNIST event uses 64 per rank.
    batch_size: int = 64
    # if set, per-rank batch = global_batch // world_size (reference CIFAR)
    global_batch: Optional[int] = None
    shuffle: bool = True          # DistributedRandomSampler vs Sequential
    augment: bool = False         # pad(4) + random flip + random crop(32)
    # synthetic dataset controls
    synthetic_train_samples: int = 50000
    synthetic_test_samples: int = 10000
    num_classes: int = 10
    synthetic_noise: float = 0.5  # class-prototype noise level
    # fraction of TRAIN labels flipped to a wrong class (test stays clean);
    # keeps accuracy in a non-saturating band for discriminative
    # matched-accuracy comparisons
    label_noise: float = 0.0


@dataclass
class RunConfig:
    """Top-level config for one training run (one mode, one model)."""

    mode: str = "event"           # cent | decent | event | spevent | serial
    model: str = "resnet18q"      # mlp | cnn1 | cnn2 | lenet5 | resnet18q |
    #                               resnet18 | resnet34 | resnet50 | ...
    epochs: int = 20
    seed: int = 0                 # torch::manual_seed(0) everywhere in ref
    topk_percent: float = 1.0     # spevent top-k % (ref argv[4])
    trigger: TriggerConfig = field(default_factory=TriggerConfig)
    optim: OptimConfig = field(default_factory=OptimConfig)
    data: DataConfig = field(default_factory=DataConfig)
    # observability (ref file_write flag, event.cpp:89 / §5)
    trace: bool = False           # write send{rank}.txt / recv{rank}.txt
    trace_dir: str = "."
    log_interval: int = 20        # dcifar10 Options::log_interval
    #                               (defined-but-unused in the reference too)
    # checkpointing (new capability; reference has none — SURVEY.md §5)
    checkpoint_path: Optional[str] = None
    checkpoint_every_epochs: int = 0   # 0 = only at end if path set
    resume: bool = False
    # device/dtype
    device: str = "auto"          # auto | cpu | cuda
    dist_backend: Optional[str] = None  # override (nccl/gloo); None = auto
    compute_dtype: str = "bf16"   # bf16 | fp32 (GPU compute dtype; params fp32)
    hip_graph: bool = True        # capture fwd+bwd into a hipGraph on GPU
    #                               (auto-skipped for models with dropout)
    # evaluation
    eval_at_end: bool = True
    final_consensus: bool = True  # closing param AllReduce (event.cpp:517-525)

    def to_json(self) -> str:
        return json.dumps(dataclasses.asdict(self), indent=2)

    @staticmethod
    def from_dict(d: dict) -> "RunConfig":
        d = dict(d)
        for key, cls in ("trigger", TriggerConfig), ("optim", OptimConfig), (
            "data",
            DataConfig,
        ):
            if key in d and isinstance(d[key], dict):
                d[key] = cls(**d[key])
        return RunConfig(**d)

    @staticmethod
    def from_json(s: str) -> "RunConfig":
        return RunConfig.from_dict(json.loads(s))


def preset(name: str) -> RunConfig:
    """Reference-experiment presets, matching the five trainer binaries."""
    if name == "dmnist-cent":
        # cent.cpp: MLP, full-shard batch, 250 epochs, lr 1e-2, random sampler
        return RunConfig(
            mode="cent", model="mlp", epochs=250,
            optim=OptimConfig(lr=1e-2),
            data=DataConfig(dataset="mnist", batch_size=0, shuffle=True),
        )
    if name == "dmnist-decent":
        # decent.cpp: MLP, full-shard batch, 50 epochs, lr 1e-2, seq sampler
        return RunConfig(
            mode="decent", model="mlp", epochs=50,
            optim=OptimConfig(lr=1e-2),
            data=DataConfig(dataset="mnist", batch_size=0, shuffle=False),
        )
    if name == "dmnist-event":
        # event.cpp: CNN-2, batch 64, 10 epochs, lr 0.05, seq sampler
        return RunConfig(
            mode="event", model="cnn2", epochs=10,
            optim=OptimConfig(lr=0.05),
            data=DataConfig(dataset="mnist", batch_size=64, shuffle=False),
        )
    if name == "dcifar10-event":
        # dcifar10/event/event.cpp: quirk-ResNet, global batch 256, 20 epochs,
        # lr 1e-2 momentum 0.9, random sampler + augmentation
        return RunConfig(
            mode="event", model="resnet18q", epochs=20,
            optim=OptimConfig(lr=1e-2, momentum=0.9),
            data=DataConfig(dataset="cifar10", global_batch=256, batch_size=256,
                            shuffle=True, augment=True),
        )
    if name == "dcifar10-spevent":
        cfg = preset("dcifar10-event")
        cfg.mode = "spevent"
        return cfg
    raise ValueError(f"unknown preset: {name!r}")


PRESETS = (
    "dmnist-cent",
    "dmnist-decent",
    "dmnist-event",
    "dcifar10-event",
    "dcifar10-spevent",
)
