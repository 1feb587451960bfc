"""Per-rank send/recv trace files, format-compatible with the reference.

The reference, under ``file_write == 1``, writes per pass one line per file
(event.cpp:232-252,337-339,385-391,418-425,482-485):

  send{rank}.txt: "<norm>,  <thres>,  <fired01>,  " per tensor
  recv{rank}.txt: per neighbor (left then right): ["1,  " if new msg]
                  "<recv_norm>,  " per tensor

We keep the same filenames and column meaning, but always write the 0/1
new-msg flag (the reference omits the '0' case on MNIST recv — ambiguous to
parse; dcifar10/spevent writes both, spevent.cpp:470-478, which is the format
used here). A train{rank}.txt carries per-epoch accuracy+loss, and
values{rank}.txt carries the reference cent binary's exact per-epoch
"<epoch>, <loss>" lines (cent.cpp:80-92,124).
"""

from __future__ import annotations

import os


class Tracer:
    def __init__(self, rank: int, outdir: str = "."):
        os.makedirs(outdir, exist_ok=True)
        self.fps = open(os.path.join(outdir, f"send{rank}.txt"), "w")
        self.fpr = open(os.path.join(outdir, f"recv{rank}.txt"), "w")
        self.fpt = open(os.path.join(outdir, f"train{rank}.txt"), "w")
        self.fpv = open(os.path.join(outdir, f"values{rank}.txt"), "w")

    def send_line(self, norms, thres, fired):
        parts = [f"{n},  {t},  {int(f)},  "
                 for n, t, f in zip(norms, thres, fired)]
        self.fps.write("".join(parts) + "\n")

    def recv_line(self, left_norms, left_new, right_norms, right_new):
        parts = [f"{int(nw)},  {n},  " for n, nw in zip(left_norms, left_new)]
        parts += [f"{int(nw)},  {n},  "
                  for n, nw in zip(right_norms, right_new)]
        self.fpr.write("".join(parts) + "\n")

    def train_line(self, epoch: int, accuracy: float, loss: float):
        self.fpt.write(f"{epoch}, {accuracy}, {loss}\n")
        self.fpv.write(f"{epoch}, {loss}\n")  # cent.cpp:124 format

    def close(self):
        for f in (self.fps, self.fpr, self.fpt, self.fpv):
            f.close()
