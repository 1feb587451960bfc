"""Communication engines for the four training modes (+ serial).

Per-iteration contract with the trainer (pass_num counts from 1):

    engine.begin_pass(pass_num)   # BEFORE forward: trigger decision, post
                                  # isend/irecv so transfer overlaps fwd+bwd
    ... forward / loss / backward ...
    engine.after_backward()       # cent: fused-grad allreduce (+ /world);
                                  # gossip: wait transfers, unpack inboxes,
                                  # theta <- (theta + left + right)/3
    engine.step()                 # fused SGD step + per-tensor L2 norms
    ...
    engine.finalize()             # closing consensus allreduce on params
                                  # (event.cpp:517-525) + num_events reduce

Algorithm equivalence with the reference (which orders: backward ->
[send theta_t; read inbox; average] -> step, event.cpp:301-488): theta does
not change between "after step t-1" and "before averaging at pass t", so
evaluating the trigger on theta_t and POSTING the sends right after step t-1
(i.e. in begin_pass(t)) transmits exactly the same values the reference's
MPI_Put sends at pass t, while letting RCCL move them during forward+backward.
The norms the trigger needs are a by-product of the fused SGD-step kernel.
"""

from __future__ import annotations

import numpy as np
import torch
import torch.distributed as dist

from ..config import RunConfig
from ..ops.backend import native
from .controller import TriggerController
from .flat import FlatParamSpace
from .transport import RingTransport


# ---------------------------------------------------------------------------
# math backends (CPU torch oracle / HIP native)
# ---------------------------------------------------------------------------

class _CpuK:
    """Plain-torch math for CPU runs and as the tested oracle."""

    @staticmethod
    def sqnorms(space: FlatParamSpace, buf: torch.Tensor) -> torch.Tensor:
        return space.sqnorms_cpu(buf)

    @staticmethod
    def sgd_step(space: FlatParamSpace, lr, momentum, wd) -> torch.Tensor:
        g = space.grad
        if wd:
            g = g.add(space.param, alpha=wd)
        if momentum:
            space.momentum.mul_(momentum).add_(g)
            upd = space.momentum
        else:
            upd = g
        space.param.add_(upd, alpha=-lr)
        return space.sqnorms_cpu(space.param)

    @staticmethod
    def avg3(param, left, right):
        # event.cpp:469-471: p += l; p += r; p /= 3
        param.add_(left).add_(right).div_(3.0)

    @staticmethod
    def pack(space, buf, fired_idx) -> torch.Tensor:
        if len(fired_idx) == 0:
            return buf.new_empty(0)
        return torch.cat([space.seg(buf, i) for i in fired_idx])

    @staticmethod
    def unpack(space, payload, fired_idx, inbox) -> None:
        off = 0
        for i in fired_idx:
            n = space.numels[i]
            space.seg(inbox, i).copy_(payload[off:off + n])
            off += n


class _GpuK:
    """HIP-native math (eventgrad_amd._core); fails loudly if unbuilt."""

    @staticmethod
    def sqnorms(space, buf):
        return native().seg_sqnorms(buf, space.starts_t, space.numels_t)

    @staticmethod
    def sgd_step(space, lr, momentum, wd):
        return native().sgd_step_norm(space.param, space.grad, space.momentum,
                                      space.starts_t, space.numels_t,
                                      float(lr), float(momentum), float(wd))

    @staticmethod
    def avg3(param, left, right):
        native().avg3(param, left, right)

    @staticmethod
    def pack(space, buf, fired_idx) -> torch.Tensor:
        if len(fired_idx) == 0:
            return buf.new_empty(0)
        dev = buf.device
        src = torch.tensor([space.starts[i] for i in fired_idx],
                           dtype=torch.int64, device=dev)
        sizes = [space.numels[i] for i in fired_idx]
        cum = np.concatenate([[0], np.cumsum(sizes)])
        offs = torch.tensor(cum, dtype=torch.int64, device=dev)
        return native().gather_segments(buf, src, offs, int(cum[-1]))

    @staticmethod
    def unpack(space, payload, fired_idx, inbox) -> None:
        if len(fired_idx) == 0:
            return
        dev = payload.device
        dst = torch.tensor([space.starts[i] for i in fired_idx],
                           dtype=torch.int64, device=dev)
        sizes = [space.numels[i] for i in fired_idx]
        offs = torch.tensor(np.concatenate([[0], np.cumsum(sizes)]),
                            dtype=torch.int64, device=dev)
        native().scatter_segments(payload, dst, offs, inbox)


def _kernels(device: torch.device):
    return _GpuK if device.type == "cuda" else _CpuK


# ---------------------------------------------------------------------------
# engines
# ---------------------------------------------------------------------------

class CommEngine:
    def __init__(self, space: FlatParamSpace, cfg: RunConfig, rank: int,
                 world: int, device: torch.device, tracer=None):
        self.space = space
        self.cfg = cfg
        self.rank, self.world = rank, world
        self.device = device
        self.K = _kernels(device)
        self.tracer = tracer
        self.pass_num = 0

    # num_events lives in the controller when one exists (device-resident
    # on GPU); engines without a trigger (serial/cent) keep a plain counter.
    @property
    def num_events(self) -> int:
        ctrl = getattr(self, "ctrl", None)
        if ctrl is not None:
            return ctrl.num_events
        return getattr(self, "_num_events", 0)

    @num_events.setter
    def num_events(self, v: int) -> None:
        ctrl = getattr(self, "ctrl", None)
        if ctrl is not None:
            ctrl.num_events = v
        else:
            self._num_events = v

    # lifecycle
    def begin_pass(self, pass_num: int) -> None:
        self.pass_num = pass_num

    def after_backward(self) -> None:
        pass

    def step(self) -> None:
        o = self.cfg.optim
        self._last_norms_sq = self.K.sgd_step(self.space, o.lr, o.momentum,
                                              o.weight_decay)
        # the step is the last param mutation of a pass: regenerate the
        # bf16 weight shadows the next forward/backward will read
        self.space.refresh_shadows()

    def drain(self) -> None:
        """Complete any in-flight lookahead comm (no-op for most engines).

        Must run before the process group is destroyed even when the
        closing consensus (finalize) is skipped (final_consensus=False)."""

    def finalize(self) -> dict:
        """Closing consensus: params <- allreduce/world (event.cpp:517-525).

        Unlike the reference (which divides on rank 0 only, cent-style), we
        divide on every rank so all ranks end with the consensus model.
        """
        if self.world > 1:
            dist.all_reduce(self.space.param)
            self.space.param.div_(self.world)
            self.space.refresh_shadows()  # post-consensus eval reads them
            # NCCL needs a device tensor (gloo accepts either)
            ev = torch.tensor([self.num_events], dtype=torch.int64,
                              device=self.device)
            dist.all_reduce(ev)
            total_events = int(ev.item())
        else:
            total_events = self.num_events
        return {"num_events_local": self.num_events,
                "num_events_total": total_events}

    # helpers
    def _norms(self) -> np.ndarray:
        """L2 norms of current params (sqrt of the step kernel's output)."""
        sq = getattr(self, "_last_norms_sq", None)
        if sq is None:
            sq = self.K.sqnorms(self.space, self.space.param)
        return np.sqrt(sq.detach().cpu().numpy().astype(np.float32))


class SerialEngine(CommEngine):
    """world==1: no communication at all (numranks>1 guard,
    dcifar10/event/event.cpp:70,281)."""


class CentEngine(CommEngine):
    """Centralized synchronous DP: allreduce the fused flat gradient then
    divide by world (cent.cpp:130-142, fused into one buffer per C1)."""

    def after_backward(self) -> None:
        if self.world > 1:
            dist.all_reduce(self.space.grad)
            if self.cfg.optim.average_grads:
                self.space.grad.div_(self.world)


class GossipEngine(CommEngine):
    """decent (always-fire) and event (norm-delta trigger) ring gossip."""

    def __init__(self, space, cfg, rank, world, device, tracer=None):
        super().__init__(space, cfg, rank, world, device, tracer)
        t = cfg.trigger
        kw = dict(adaptive=t.adaptive, horizon=t.horizon,
                  constant=t.constant, sent_history=t.sent_history,
                  initial_comm_passes=t.initial_comm_passes,
                  always_fire=(cfg.mode == "decent"))
        if device.type == "cuda":
            from .controller import GpuTriggerController
            self.ctrl = GpuTriggerController(space.sz, device, **kw)
        else:
            self.ctrl = TriggerController(space.sz, **kw)
        self.transport = (RingTransport(rank, world, device)
                          if world > 1 else None)
        self.inbox_left = space.new_like()
        self.inbox_right = space.new_like()
        self._recv_l = None
        self._recv_r = None
        self._fired_l: list = []
        self._fired_r: list = []
        # lookahead: (pass_num, fire mask) decided at the end of the
        # previous step(), whose mask exchange is already in flight
        self._pending: tuple | None = None

    # -- send-side payload construction (dense) --------------------------
    def _make_send(self, fired_idx):
        payload = self.K.pack(self.space, self.space.param, fired_idx)
        # two independent buffers so the two isends never alias
        return payload, payload.clone() if payload.numel() else payload

    def _make_recv(self, fired_idx):
        n = sum(self.space.numels[i] for i in fired_idx)
        return torch.empty(n, dtype=torch.float32, device=self.device)

    def _apply_recv(self, payload, fired_idx, inbox):
        self.K.unpack(self.space, payload, fired_idx, inbox)

    def _sq(self):
        sq = getattr(self, "_last_norms_sq", None)
        if sq is None:
            sq = self.K.sqnorms(self.space, self.space.param)
        return sq

    def _all_fire_static(self, pass_num: int) -> bool:
        """True when the fire decision is deterministically all-ones on
        EVERY rank (decent mode / warmup) — no mask exchange needed."""
        return (self.ctrl.always_fire
                or pass_num < self.ctrl.initial_comm_passes)

    def step(self) -> None:
        super().step()
        # lookahead: decide pass+1's trigger NOW (pure, no state commit)
        # and post the tiny mask exchange so its wire time overlaps the
        # host work between passes (accuracy sync, batch prep) instead of
        # blocking at the top of begin_pass. The commit happens in
        # begin_pass; a decision whose pass never runs commits nothing.
        if self.world <= 1:
            return
        nxt = self.pass_num + 1
        if self._all_fire_static(nxt):
            return  # begin_pass takes the static fast path, no exchange
        if self.device.type == "cuda":
            fire = self.ctrl.decide_device(self._last_norms_sq, nxt)
        else:
            fire = self.ctrl.decide(self._norms(), nxt)
        self.transport.post_masks(torch.from_numpy(fire.astype(np.uint8)))
        self._pending = (nxt, fire)

    def begin_pass(self, pass_num: int) -> None:
        self.pass_num = pass_num
        if self.world <= 1:
            return
        pending = self._pending
        self._pending = None
        have_lookahead = pending is not None and pending[0] == pass_num
        if have_lookahead:
            # commit the controller state for the pre-decided mask; the
            # commit recomputes the identical decision from the same
            # (unchanged) state + norms, so no second mask D2H is needed
            fire = pending[1]
            if self.device.type == "cuda":
                self.ctrl.step_device(self._sq(), pass_num, need_mask=False)
                self.ctrl.last_fired = fire
            else:
                committed = self.ctrl.step(self._norms(), pass_num)
                assert (committed == fire).all()
        elif self.device.type == "cuda":
            # device-resident controller: only the mask crosses to host
            fire = self.ctrl.step_device(self._sq(), pass_num)
        else:
            fire = self.ctrl.step(self._norms(), pass_num)
        if self.tracer is not None:
            if self.device.type == "cuda":
                norms, thres = self.ctrl.trace_values()
            else:
                norms, thres = self.ctrl.last_norms, self.ctrl.thres
            self.tracer.send_line(norms, thres, fire)
        my_fired = [i for i in range(self.space.sz) if fire[i]]
        if self._all_fire_static(pass_num):
            # all-ones on every rank: skip the mask exchange entirely
            assert fire.all()
            self._fired_l = list(range(self.space.sz))
            self._fired_r = list(range(self.space.sz))
        else:
            if have_lookahead:
                mask_l, mask_r = self.transport.wait_masks()
            else:  # first pass / post-resume fallback: blocking exchange
                mask = torch.from_numpy(fire.astype(np.uint8))
                mask_l, mask_r = self.transport.exchange_masks(mask)
            self._fired_l = [i for i in range(self.space.sz)
                             if int(mask_l[i])]  # left neighbor fired
            self._fired_r = [i for i in range(self.space.sz)
                             if int(mask_r[i])]
        send_l, send_r = self._make_send(my_fired)
        self._recv_l = self._make_recv(self._fired_l)
        self._recv_r = self._make_recv(self._fired_r)
        self.transport.post_payloads(send_l, send_r, self._recv_l,
                                     self._recv_r)

    def after_backward(self) -> None:
        if self.world <= 1:
            return
        self.transport.finish()
        self._apply_recv(self._recv_l, self._fired_l, self.inbox_left)
        self._apply_recv(self._recv_r, self._fired_r, self.inbox_right)
        if self.tracer is not None:
            self._trace_recv()
        self.K.avg3(self.space.param, self.inbox_left, self.inbox_right)

    def drain(self) -> None:
        # drain a lookahead mask exchange whose pass never ran (training
        # ended): peers posted symmetrically, so waiting completes it
        if self.transport is not None:
            self.transport.cancel_pending_masks()
        self._pending = None

    def finalize(self) -> dict:
        self.drain()
        return super().finalize()

    def _trace_recv(self):
        # reference logs, per tensor: new-msg flag + received-half norm
        # (event.cpp:399-461). New-msg is exact here (the neighbor's mask).
        lns, rns = [], []
        for i in range(self.space.sz):
            lns.append(float(self.space.seg(self.inbox_left, i).norm()))
            rns.append(float(self.space.seg(self.inbox_right, i).norm()))
        newl = [i in set(self._fired_l) for i in range(self.space.sz)]
        newr = [i in set(self._fired_r) for i in range(self.space.sz)]
        self.tracer.recv_line(lns, newl, rns, newr)


class SparseGossipEngine(GossipEngine):
    """spevent: event trigger + top-k sparsified messages
    (dcifar10/spevent/spevent.cpp:309-551).

    Wire format per direction: for each FIRED tensor i, k_i fp32 values
    followed by k_i int32 indices BITCAST to fp32 (the reference ships
    indices value-converted to float, spevent.cpp:351 — a precision hazard
    above 2^24 elements; bitcasting is lossless and wire-size identical).

    Persistent state: prev (last-sent values at element granularity) and
    dense left/right neighbor replicas; averaging uses the dense replicas
    (spevent.cpp:540-542). Replicas and prev are initialised to the model's
    initial parameters — with the shared seed every rank starts identical,
    so this equals the neighbors' true initial state (the reference instead
    initialises them to fresh random models, spevent.cpp:128-136, which is
    a strictly worse approximation of the neighbor).
    """

    def __init__(self, space, cfg, rank, world, device, tracer=None):
        super().__init__(space, cfg, rank, world, device, tracer)
        self.prev = space.clone_params()
        self.inbox_left = space.clone_params()   # dense left replica
        self.inbox_right = space.clone_params()  # dense right replica
        pct = cfg.topk_percent / 100.0
        self.k = [max(1, int(np.ceil(pct * n))) for n in space.numels]

    def _payload_elems(self, fired_idx):
        return sum(2 * self.k[i] for i in fired_idx)

    def _seg_tables(self, fired_idx):
        """Device tables for the batched pack/unpack kernels.

        offs layout: [val_offs[0..nf-1], total_payload,
                      cum_k[0..nf-1], total_k]  (see csrc/topk.hip).
        """
        dev = self.device
        val_offs, cum_k = [], []
        off = ck = 0
        for i in fired_idx:
            val_offs.append(off)
            cum_k.append(ck)
            off += 2 * self.k[i]
            ck += self.k[i]
        starts = torch.tensor([self.space.starts[i] for i in fired_idx],
                              dtype=torch.int64, device=dev)
        lens = torch.tensor([self.space.numels[i] for i in fired_idx],
                            dtype=torch.int64, device=dev)
        ks = torch.tensor([self.k[i] for i in fired_idx],
                          dtype=torch.int64, device=dev)
        offs = torch.tensor(val_offs + [off] + cum_k + [ck],
                            dtype=torch.int64, device=dev)
        return starts, lens, ks, offs, off, ck

    def _make_send(self, fired_idx):
        if not fired_idx:
            e = self.space.param.new_empty(0)
            return e, e
        if self.device.type == "cuda":
            # one fused 9-launch radix-select for ALL fired tensors; the
            # kernel writes the wire payload and updates prev in place
            starts, lens, ks, offs, total, _ = self._seg_tables(fired_idx)
            payload = native().spevent_pack(
                self.space.param, self.prev, starts, lens, ks, offs, total,
                max(self.space.numels[i] for i in fired_idx))
            return payload, payload.clone()
        chunks = []
        for i in fired_idx:
            seg = self.space.seg(self.space.param, i)
            pseg = self.space.seg(self.prev, i)
            diff = (seg - pseg).abs()
            _, idx = torch.topk(diff, self.k[i], sorted=True)
            vals = seg[idx]
            # update prev at the sent indices only (spevent.cpp:407-413)
            pseg[idx] = vals
            idx = idx.to(torch.int32)
            chunks.append(vals.to(torch.float32))
            chunks.append(idx.view(torch.float32))
        payload = torch.cat(chunks)
        return payload, payload.clone()

    def _make_recv(self, fired_idx):
        return torch.empty(self._payload_elems(fired_idx),
                           dtype=torch.float32, device=self.device)

    def _apply_recv(self, payload, fired_idx, replica):
        if not fired_idx:
            return
        if self.device.type == "cuda":
            starts, lens, ks, offs, _, total_k = self._seg_tables(fired_idx)
            native().spevent_unpack(payload, starts, ks, offs, replica,
                                    total_k)
            return
        off = 0
        for i in fired_idx:
            k = self.k[i]
            vals = payload[off:off + k]
            idx = payload[off + k:off + 2 * k].view(torch.int32)
            self.space.seg(replica, i)[idx.long()] = vals
            off += 2 * k


def build_engine(space: FlatParamSpace, cfg: RunConfig, rank: int, world: int,
                 device: torch.device, tracer=None) -> CommEngine:
    mode = cfg.mode
    if world <= 1 and mode != "serial":
        # reference guard: single rank trains serially (event.cpp:281 guard)
        mode = "serial"
    cls = {
        "serial": SerialEngine,
        "cent": CentEngine,
        "decent": GossipEngine,
        "event": GossipEngine,
        "spevent": SparseGossipEngine,
    }[mode]
    return cls(space, cfg, rank, world, device, tracer)
