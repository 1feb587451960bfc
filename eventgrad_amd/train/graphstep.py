"""hipGraph capture of the forward+backward region.

The per-step hot path launches ~300 kernels; on MI355X the CPU-side launch
gaps plus the small aten glue (autograd grad-accumulate adds, weight
transforms, workspace fills) cost ~1 ms/step. Capturing
[zero_grad -> forward -> loss -> backward] into one hipGraph replays them
with no per-kernel CPU involvement. Communication (mask exchange, payload
waits, averaging) and the fused SGD step stay eager — they are identical
at every world size, so graphing does not skew the scaling curve's
compute region.

Requirements (all true for the flagship ResNet path):
  * static batch shape (ShardLoader drop_last / bench resident batches);
  * all grads pre-allocated as views of the flat grad buffer (they are —
    FlatParamSpace re-points p.grad);
  * no host syncs inside fwd/bwd (loss backward reads its upstream grad
    as a device scalar);
  * no per-step host RNG (models with dropout set `uses_dropout` and are
    not captured).
"""

from __future__ import annotations

import torch

from ..ops import functional as O


def can_graph(model, device) -> bool:
    if device.type != "cuda":
        return False
    if getattr(model, "uses_dropout", False):
        return False
    # fp32 mode runs aten/MIOpen ops; capture of library convs (workspace
    # allocation, autotuning) is not graph-safe — eager only.
    if O.get_compute_dtype() != "bf16":
        return False
    return True


class FwdBwdGraph:
    def __init__(self, model, space, x_shape, device,
                 warmup_iters: int = 3):
        self.model = model
        self.space = space
        self.device = device
        self.static_x = torch.zeros(x_shape, device=device)
        self.static_y = torch.zeros(x_shape[0], dtype=torch.long,
                                    device=device)
        self._graph = None
        self._warmup = warmup_iters

    def _run_eager(self):
        self.space.zero_grad()
        logits = self.model(self.static_x)
        loss = O.nll_of_logits(logits, self.static_y)
        loss.backward()
        return logits, loss

    def _capture(self):
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(self._warmup):
                self._run_eager()
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self.static_logits, self.static_loss = self._run_eager()
        self._graph = g

    def step(self, x, y):
        """Copy the batch into the static buffers and replay fwd+bwd.

        Returns (logits, loss) — STATIC tensors, valid until next step.
        Falls back to eager execution permanently if capture fails.
        """
        self.static_x.copy_(x, non_blocking=True)
        self.static_y.copy_(y, non_blocking=True)
        if self._graph is None:
            try:
                self._capture()
            except Exception as e:  # pragma: no cover - defensive
                import warnings
                warnings.warn(f"hipGraph capture failed ({e!r}); "
                              "falling back to eager execution")
                self._graph = False
        if self._graph is False:
            self.static_logits, self.static_loss = self._run_eager()
        else:
            self._graph.replay()
        return self.static_logits, self.static_loss
