"""Process bootstrap & communicator init (L5/L0 of SURVEY.md §1).

One process per GPU over torch.distributed; backend "nccl" IS RCCL on ROCm
(xGMI intra-node). Env contract matches torchrun:
RANK/WORLD_SIZE/LOCAL_RANK/MASTER_ADDR/MASTER_PORT
(/root/reference/02-distributed-data-parallel/train_llm.py:36-41).

Also provides the barrier-fenced orderings the reference uses:
rank0_first (02:272-280) and rank_ordered (06:346-353).
"""
import datetime
import os
from contextlib import contextmanager

import torch
import torch.distributed as dist


def _env_int(*names, default):
    for n in names:
        v = os.environ.get(n)
        if v is not None:
            return int(v)
    return default


def env_rank() -> int:
    # torchrun sets RANK; under mpirun OpenMPI sets OMPI_COMM_WORLD_RANK
    # (the reference's mpi variant reads the same, 03-.../README.md:127-133)
    return _env_int("RANK", "OMPI_COMM_WORLD_RANK", default=0)


def env_world_size() -> int:
    return _env_int("WORLD_SIZE", "OMPI_COMM_WORLD_SIZE", default=1)


def env_local_rank() -> int:
    return _env_int("LOCAL_RANK", "OMPI_COMM_WORLD_LOCAL_RANK", default=0)


def init_distributed(device: torch.device | None = None,
                     timeout_s: int = 600) -> tuple[int, int, int]:
    """init_process_group with the right backend for the device; returns
    (rank, local_rank, world_size)."""
    rank, local_rank, world = env_rank(), env_local_rank(), env_world_size()
    if dist.is_initialized():
        return dist.get_rank(), local_rank, dist.get_world_size()
    if "RANK" not in os.environ and "OMPI_COMM_WORLD_RANK" in os.environ:
        # let torch's env:// rendezvous see the mpi-provided identity
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ.setdefault("LOCAL_RANK", str(local_rank))
    if world == 1 and "MASTER_ADDR" not in os.environ:
        # single-process launch without torchrun: self-rendezvous
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
    backend = "nccl" if (device is not None and device.type == "cuda") else \
        ("nccl" if torch.cuda.is_available() else "gloo")
    kwargs = dict(rank=rank, world_size=world,
                  timeout=datetime.timedelta(seconds=timeout_s))
    if backend == "nccl" and device is not None:
        kwargs["device_id"] = device
    dist.init_process_group(backend, **kwargs)
    return rank, local_rank, world


@contextmanager
def rank0_first():
    """Rank 0 runs the body first, everyone else after (barrier x2)."""
    rank = dist.get_rank() if dist.is_initialized() else 0
    if rank == 0:
        yield
    if dist.is_initialized():
        dist.barrier()
    if rank != 0:
        yield
    if dist.is_initialized():
        dist.barrier()


@contextmanager
def rank_ordered(should_go_first: bool):
    """`should_go_first` ranks run the body before the rest (06:346-353)."""
    if should_go_first:
        yield
    if dist.is_initialized():
        dist.barrier()
    if not should_go_first:
        yield
    if dist.is_initialized():
        dist.barrier()


def destroy():
    if dist.is_initialized():
        dist.destroy_process_group()
