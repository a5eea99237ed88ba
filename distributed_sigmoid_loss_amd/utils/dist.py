"""Process-group lifecycle helpers for tests and single-node launches.

Parity: reference ``setup``/``cleanup`` (``test_distributed_sigmoid_loss.py:35-54``),
modernized: rendezvous always on 127.0.0.1 (container hostnames may not
resolve), backend selected for the hardware (RCCL via the "nccl" backend name
on ROCm when a GPU is present, gloo otherwise).
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def setup_process_group(rank: int, world_size: int, backend: str = "gloo",
                        port: int = 29571, timeout_s: int = 300) -> None:
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", str(port))
    dist.init_process_group(backend, rank=rank, world_size=world_size,
                            timeout=datetime.timedelta(seconds=timeout_s))


def cleanup_process_group() -> None:
    if dist.is_available() and dist.is_initialized():
        dist.destroy_process_group()


def auto_backend() -> str:
    return "nccl" if torch.cuda.is_available() else "gloo"


def init_from_env() -> tuple[int, int, int]:
    """torchrun-style init: read RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* from env.

    Returns (rank, local_rank, world_size).  Binds the process to its GPU
    before init so RCCL communicators land on the right device.
    """
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        backend = auto_backend()
        if backend == "nccl":
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend, rank=rank, world_size=world)
    elif torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return rank, local_rank, world
