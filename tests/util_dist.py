"""Helpers for multi-process (gloo, world_size>1) CPU tests."""

from __future__ import annotations

import os

import torch.multiprocessing as mp

_NEXT_PORT = [29100 + (os.getpid() % 100) * 3]  # disjoint from benchmarks (29650+)


def next_port() -> int:
    _NEXT_PORT[0] += 3  # leave room for per-cycle +1/+2 offsets in workers
    return _NEXT_PORT[0]


def run_world(worker, world: int, *args, port: int | None = None) -> None:
    """Spawn `world` processes running worker(rank, world, port, *args)."""
    port = port or next_port()
    mp.start_processes(worker, args=(world, port) + tuple(args),
                       nprocs=world, start_method="spawn", join=True)


def init_env(rank: int, world: int, port: int) -> None:
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
