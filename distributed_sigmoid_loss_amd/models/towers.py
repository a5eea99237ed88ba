"""Two-tower encoder stand-ins for tests and the benchmark.

The reference's "model" is a pair of identically-seeded bias-free
``nn.Linear(emb_dim, 2)`` towers (``test_distributed_sigmoid_loss.py:71-76``)
— small enough that encoder gradients are exact probes of the loss.  We keep
that toy for the oracles and add a projection tower of realistic width for the
benchmark (so the bench step is a real train step: encode → normalize → loss →
backward → optimizer).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


def make_toy_towers(emb_dim: int, out_dim: int = 2, seed: int = 3407):
    """Identically-initialized bias-free Linear image/text towers.

    Reseeding before each construction makes every rank build the same
    weights (the reference's trick, ``test_distributed_sigmoid_loss.py:71-76``).
    """
    torch.manual_seed(seed)
    image_encoder = nn.Linear(emb_dim, out_dim, bias=False)
    torch.manual_seed(seed)
    text_encoder = nn.Linear(emb_dim, out_dim, bias=False)
    return image_encoder, text_encoder


class ProjectionTower(nn.Module):
    """Single-layer projection tower ``in_dim → emb_dim`` with L2 output norm.

    Used by the benchmark as the flagship encoder: big enough to exercise a
    real optimizer step and DDP gradient averaging, small enough that the
    contrastive loss (quadratic in batch) dominates — which is what this
    framework accelerates.
    """

    def __init__(self, in_dim: int, emb_dim: int):
        super().__init__()
        self.proj = nn.Linear(in_dim, emb_dim, bias=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from ..ops import l2_normalize
        return l2_normalize(self.proj(x))


class TwoTowerModel(nn.Module):
    """Image + text projection towers sharing no weights.

    (Round 2 negative result, banked: batching the two skinny projections
    into one strided-batched GEMM via stacked weights measured +0.15 ms —
    the input-stack copies outweigh the batched-GEMM win at this shape.
    Two plain GEMMs + the fused normalize kernels it is.)
    """

    def __init__(self, in_dim: int, emb_dim: int):
        super().__init__()
        self.image = ProjectionTower(in_dim, emb_dim)
        self.text = ProjectionTower(in_dim, emb_dim)

    def forward(self, image_feats: torch.Tensor, text_feats: torch.Tensor):
        return self.image(image_feats), self.text(text_feats)
