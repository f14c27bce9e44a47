"""Helpers for multi-process CPU (gloo) tests."""

import os
import socket

import torch.distributed as dist
import torch.multiprocessing as mp


def free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _entry(rank, world_size, port, fn, args):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        fn(rank, world_size, *args)
    finally:
        dist.destroy_process_group()


def run_dist(fn, world_size=2, args=(), timeout=180):
    """Spawn `world_size` gloo processes running fn(rank, world_size, *args)."""
    port = free_port()
    mp.spawn(
        _entry,
        args=(world_size, port, fn, args),
        nprocs=world_size,
        join=True,
    )
