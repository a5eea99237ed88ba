"""Determinism helpers (parity: reference ``test_distributed_sigmoid_loss.py:15-32``)."""

from __future__ import annotations

import os
import random

import numpy as np
import torch


def set_seed(seed: int, deterministic_cudnn: bool = False) -> None:
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
    np.random.seed(seed)
    random.seed(seed)
    if deterministic_cudnn:
        torch.backends.cudnn.deterministic = True
        torch.backends.cudnn.benchmark = False


def seeded_global_batch(global_batch: int, dim: int, seed: int,
                        dtype=torch.float32) -> torch.Tensor:
    """Draw the *global* batch with a fixed seed — every rank sees the same
    data and slices its own shard (the reference's multi-rank-equals-1-rank
    mechanism, ``test_distributed_sigmoid_loss.py:57-68``)."""
    g = torch.Generator().manual_seed(seed)
    return torch.randn(global_batch, dim, generator=g, dtype=dtype)


def rank_shard(x: torch.Tensor, rank: int, world_size: int) -> torch.Tensor:
    b = x.shape[0] // world_size
    return x[rank * b:(rank + 1) * b]
