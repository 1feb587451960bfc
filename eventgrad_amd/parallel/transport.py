"""Ring p2p transport: event-triggered neighbor exchange over RCCL/xGMI.

Replaces the reference's one-sided MPI RMA (MPI_Win_create/MPI_Put,
event.cpp:169-179,348-360). RCCL send/recv must be matched, so EventGraD's
unmatched Put becomes a two-phase protocol:

  phase A: exchange the per-tensor fire MASK with both neighbors
           (sz bytes — tiny);
  phase B: post matched isend/irecv for ONE fused payload per direction
           containing only the fired tensors (tight-packed fp32), sized
           from the masks on both sides; zero-size payloads are skipped.

Phase B is posted before the next forward so the transfer overlaps
forward+backward (RCCL runs on its own HIP stream); ``finish()`` is called
after backward, before the neighbor averaging, so the inbox deterministically
holds this pass's fired segments and the last received values for all others
("hold last value" semantics, event.cpp:399-456 — made deterministic: stale
is by algorithm, never by racing transfers).

Message-direction convention: a rank's send "to left" lands in the left
neighbor's FROM-RIGHT inbox (the reference writes it at window offset
num_elem_param+disp — the right half, event.cpp:348-353). NCCL has no tags;
per-peer FIFO ordering disambiguates world_size==2 (both neighbors are the
same peer): every rank posts [send-to-left, send-to-right, recv-from-right,
recv-from-left] so sender's k-th message to a peer matches that peer's k-th
posted recv. gloo uses explicit tags instead.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from .dist import ring_neighbors

TAG_L = 2    # "L-direction" message (to left / from right), ref LTAG decent.cpp:5
TAG_R = 10   # "R-direction" message (to right / from left), ref RTAG decent.cpp:6
TAG_MASK = 100
TAG_PAYLOAD = 200


class RingTransport:
    def __init__(self, rank: int, world: int, device: torch.device):
        self.rank, self.world = rank, world
        self.device = device
        self.left, self.right = ring_neighbors(rank, world)
        self.backend = dist.get_backend() if dist.is_initialized() else None
        self.use_tags = self.backend == "gloo"
        self._reqs: List = []
        self._recv_copies: List = []  # (host_staging, device_dst) for gloo+GPU
        self._mask_state = None       # in-flight post_masks state

    def _ops(self, pairs):
        """pairs: list of (kind, tensor, peer, tag). Returns work handles.

        gloo p2p requires CPU tensors; when the payload lives on GPU (the
        single-GPU multi-process test configuration) sends are staged
        through pinned host copies and recvs staged back in finish().
        """
        if self.use_tags:
            reqs = []
            for kind, t, peer, tag in pairs:
                if t.is_cuda:
                    if kind == "send":
                        t = t.cpu()
                    else:
                        host = torch.empty_like(t, device="cpu")
                        self._recv_copies.append((host, t))
                        t = host
                fn = dist.isend if kind == "send" else dist.irecv
                reqs.append(fn(t, peer, tag=tag))
            return reqs
        ops = [dist.P2POp(dist.isend if kind == "send" else dist.irecv,
                          t, peer) for kind, t, peer, _ in pairs]
        return dist.batch_isend_irecv(ops)

    # ------------------------------------------------------------------
    # mask exchange: split post/wait so the tiny phase-A transfer can be
    # posted at the END of the previous pass's optimizer step and overlap
    # the host-side work between passes (accuracy sync, next-batch prep) —
    # "move the per-pass blocking mask exchange off the host
    # critical path".
    def post_masks(self, mask: torch.Tensor) -> None:
        """Post the nonblocking mask exchange (uint8[sz]) with both neighbors.

        Must be matched by wait_masks() before the next post. The mask is
        kept referenced until wait_masks so the send buffer stays alive.
        """
        m = mask.to(self.device, dtype=torch.uint8).contiguous()
        fr_right = torch.empty_like(m)
        fr_left = torch.empty_like(m)
        pairs = [
            ("send", m, self.left, TAG_MASK + TAG_L),
            ("send", m, self.right, TAG_MASK + TAG_R),
            ("recv", fr_right, self.right, TAG_MASK + TAG_L),
            ("recv", fr_left, self.left, TAG_MASK + TAG_R),
        ]
        # keep any in-flight payload staging pairs separate from the mask's
        saved, self._recv_copies = self._recv_copies, []
        reqs = self._ops(pairs)
        self._mask_state = (reqs, m, fr_left, fr_right, self._recv_copies)
        self._recv_copies = saved

    def wait_masks(self):
        """Complete a posted mask exchange.

        Returns (mask_from_left, mask_from_right) as uint8 CPU tensors.
        """
        reqs, _m, fr_left, fr_right, copies = self._mask_state
        self._mask_state = None
        for r in reqs:
            r.wait()
        for host, dev in copies:
            dev.copy_(host)
        return fr_left.cpu(), fr_right.cpu()

    def cancel_pending_masks(self) -> None:
        """Drain a posted mask exchange whose pass never ran (end of
        training): the peers posted symmetrically, so the wire completes —
        just wait and discard before the closing collectives."""
        if getattr(self, "_mask_state", None) is not None:
            self.wait_masks()

    def exchange_masks(self, mask: torch.Tensor):
        """Blocking mask exchange (fallback when no lookahead was posted)."""
        self.post_masks(mask)
        return self.wait_masks()

    def post_payloads(self, send_l: Optional[torch.Tensor],
                      send_r: Optional[torch.Tensor],
                      recv_l: Optional[torch.Tensor],
                      recv_r: Optional[torch.Tensor]) -> None:
        """Post matched isend/irecv for the fused payloads (may be None).

        send_l/send_r: payload this rank fires to left/right (same content;
        two buffers so transfers are independent). recv_l/recv_r: buffers
        for the left/right neighbor's payload (sized by their masks).
        """
        pairs = []
        if send_l is not None and send_l.numel():
            pairs.append(("send", send_l, self.left, TAG_PAYLOAD + TAG_L))
        if send_r is not None and send_r.numel():
            pairs.append(("send", send_r, self.right, TAG_PAYLOAD + TAG_R))
        if recv_r is not None and recv_r.numel():
            pairs.append(("recv", recv_r, self.right, TAG_PAYLOAD + TAG_L))
        if recv_l is not None and recv_l.numel():
            pairs.append(("recv", recv_l, self.left, TAG_PAYLOAD + TAG_R))
        self._reqs = self._ops(pairs) if pairs else []

    def finish(self) -> None:
        for r in self._reqs:
            r.wait()
        self._reqs = []
        for host, dev in self._recv_copies:
            dev.copy_(host, non_blocking=True)
        self._recv_copies = []
