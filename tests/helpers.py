"""Shared multi-process CPU test harness.

Re-derivation of the reference's oracle mechanism
(``test_distributed_sigmoid_loss.py:35-119``): spawn ``world_size`` processes,
rendezvous over gloo on 127.0.0.1, give every rank the same seeded *global*
batch and identically-seeded toy towers, and return rank-0 results through a
manager dict.
"""

from __future__ import annotations

import socket

import torch
import torch.distributed as dist
import torch.multiprocessing as mp
import torch.nn.functional as F

from distributed_sigmoid_loss_amd.models import make_toy_towers
from distributed_sigmoid_loss_amd.utils import (
    seeded_global_batch,
    rank_shard,
    setup_process_group,
    cleanup_process_group,
)

IMG_SEED = 42  # reference draws images with seed 42, texts with 40
TXT_SEED = 40


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def encode_shard(rank: int, world: int, batch_per_rank: int, emb_dim: int,
                 out_dim: int = 2):
    """Every rank: same global batch, same towers, own contiguous shard."""
    images = seeded_global_batch(world * batch_per_rank, emb_dim, IMG_SEED)
    texts = seeded_global_batch(world * batch_per_rank, emb_dim, TXT_SEED)
    img_enc, txt_enc = make_toy_towers(emb_dim, out_dim)
    zi = F.normalize(img_enc(rank_shard(images, rank, world)), dim=-1)
    zt = F.normalize(txt_enc(rank_shard(texts, rank, world)), dim=-1)
    return img_enc, txt_enc, zi, zt


def _worker(rank, world, port, fn, args, ret):
    setup_process_group(rank, world, "gloo", port=port)
    try:
        out = fn(rank, world, *args)
        if out is not None:
            ret[rank] = out
    finally:
        cleanup_process_group()


def run_distributed(fn, world: int, *args) -> dict:
    """Run ``fn(rank, world, *args)`` on ``world`` gloo-connected processes;
    returns {rank: result} for ranks that returned non-None."""
    manager = mp.Manager()
    ret = manager.dict()
    mp.spawn(_worker, args=(world, free_port(), fn, args, ret), nprocs=world,
             join=True)
    return dict(ret)
