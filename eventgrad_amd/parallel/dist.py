"""Process-group bootstrap and ring topology.

Replaces the reference's MPI_Init/Comm_size/Comm_rank + ring arithmetic
(dmnist/event/event.cpp:108-123, dmnist/decent/decent.cpp:56-64).
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def world_info():
    """(rank, world_size, local_rank) from the torchrun/env contract."""
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local = int(os.environ.get("LOCAL_RANK", str(rank)))
    return rank, world, local


def init_distributed(device: str = "auto", backend: str | None = None,
                     timeout_s: int = 600):
    """Initialise torch.distributed (if WORLD_SIZE > 1) and pick the device.

    Returns (rank, world, torch.device). backend defaults to "nccl" (RCCL)
    on GPU and "gloo" on CPU. Rendezvous must use 127.0.0.1 in this
    environment (container hostnames may not resolve).
    """
    rank, world, local = world_info()
    use_cuda = (device in ("auto", "cuda")) and torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local % max(torch.cuda.device_count(), 1))
        dev = torch.device("cuda", torch.cuda.current_device())
    else:
        dev = torch.device("cpu")
    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if use_cuda else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s))
    return rank, world, dev


def ring_neighbors(rank: int, world: int):
    """(left, right) as in decent.cpp:56-64 / event.cpp:113-123."""
    left = world - 1 if rank == 0 else rank - 1
    right = 0 if rank == world - 1 else rank + 1
    return left, right


def barrier():
    if dist.is_initialized():
        dist.barrier()


def is_distributed() -> bool:
    return dist.is_initialized()
