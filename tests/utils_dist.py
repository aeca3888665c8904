"""Helpers for multi-process CPU (gloo) tests — world_size > 1 on one box,
the torchrun-uniform-nodes emulation the survey prescribes."""
import os
import socket

import torch.distributed as dist
import torch.multiprocessing as mp


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _entry(rank, world, port, fn, args):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        fn(rank, world, *args)
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def run_dist(fn, world_size=2, args=()):
    """Spawn `world_size` CPU processes running fn(rank, world, *args)."""
    port = free_port()
    mp.spawn(_entry, args=(world_size, port, fn, args), nprocs=world_size,
             join=True)
