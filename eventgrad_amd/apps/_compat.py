"""Shared positional-argv shim for the reference-binary entry points.

Reference contract (dmnist/event/event.cpp:89-100, spevent.cpp:45-60):
  argv[1] = file_write (0/1)
  argv[2] = thres_type (0 static / 1 adaptive)     [event/spevent only]
  argv[3] = constant (static) or horizon (adaptive)
  argv[4] = topk_percent                           [spevent only]
"""

from __future__ import annotations

import json
import sys

from ..config import preset
from ..train.trainer import Trainer


def run(preset_name: str, argv=None, with_trigger_args=True,
        with_topk=False) -> int:
    argv = list(sys.argv[1:] if argv is None else argv)
    cfg = preset(preset_name)
    if argv:
        cfg.trace = bool(int(argv[0]))
    if with_trigger_args and len(argv) >= 3:
        thres_type = int(argv[1])
        val = float(argv[2])
        if thres_type == 1:
            cfg.trigger.adaptive = True
            cfg.trigger.horizon = val
        else:
            cfg.trigger.adaptive = False
            cfg.trigger.constant = val
    if with_topk and len(argv) >= 4:
        cfg.topk_percent = float(argv[3])
    tr = Trainer(cfg)
    m = tr.train()
    if tr.rank == 0:
        print(json.dumps(m.summary()), flush=True)
    return 0
