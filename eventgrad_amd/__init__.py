"""eventgrad-mi355x: MI355X-native event-triggered decentralized SGD framework.

A brand-new framework with the capabilities of EventGraD
(soumyadipghosh/eventgrad): data-parallel training in four modes

  - ``cent``    centralized synchronous DP (AllReduce on gradients)
                [ref: dmnist/cent/cent.cpp]
  - ``decent``  decentralized ring gossip, always-communicate
                [ref: dmnist/decent/decent.cpp]
  - ``event``   EventGraD: event-triggered ring gossip with per-parameter
                norm-delta trigger + adaptive threshold controller
                [ref: dmnist/event/event.cpp, dcifar10/event/event.cpp]
  - ``spevent`` EventGraD + top-k sparsified messages
                [ref: dcifar10/spevent/spevent.cpp]

Designed MI355X-first rather than ported: the compute path is PyTorch-ROCm +
hand-written HIP/CDNA4 kernels (gfx950) + RCCL over xGMI (via
``torch.distributed`` backend "nccl", one process per GPU). The per-parameter
L2-norm used by the trigger is fused into the HIP SGD-step kernel, the
trigger/adaptive-threshold controller runs device-resident, and ring-neighbor
exchange is posted asynchronously so it overlaps the next iteration's
forward/backward.
"""

__version__ = "0.1.0"

from . import config  # noqa: F401

__all__ = ["config", "__version__"]
