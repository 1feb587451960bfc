"""Distributed layer: RCCL-over-xGMI ring gossip + collectives.

One process per GPU; wire transfer goes through ``torch.distributed``
(backend "nccl" IS RCCL on ROCm — xGMI p2p between GPUs of one node), or
"gloo" for CPU multi-process tests. The device-side work (fused SGD +
per-tensor L2 norms, trigger controller, pack/unpack, (p+l+r)/3 averaging,
top-k sparsification) runs in hand-written HIP kernels from
``eventgrad_amd._core``.

Mapping from the reference's MPI call sites:
  C1 MPI_Allreduce(grads)       -> all_reduce on ONE fused flat grad buffer
  C2-C4 Issend/Recv ring        -> batch_isend_irecv of fused payloads
  C5 RMA window (2N floats)     -> persistent device inbox flat buffers
  C6 MPI_Put on trigger         -> mask pre-exchange + matched isend/irecv,
                                   posted before forward so transfer overlaps
                                   forward+backward
  C7 stale window reads         -> inbox retains last received segment values
  C8 final param Allreduce      -> all_reduce flat param buffer
  C9 Allreduce(num_events)      -> all_reduce scalar
  C10 sparse (val,idx) Puts     -> packed (fp32 vals, bitcast int32 idx) wire
"""

from .dist import init_distributed, ring_neighbors, world_info  # noqa: F401
from .flat import FlatParamSpace  # noqa: F401
from .controller import TriggerController  # noqa: F401
from .engine import build_engine  # noqa: F401
