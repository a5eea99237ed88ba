"""distributed_sigmoid_loss_amd — MI355X-native distributed SigLIP loss framework.

A from-scratch, MI355X-first (gfx950 / CDNA4) implementation of distributed
sigmoid contrastive loss (SigLIP, arXiv:2303.15343) with the same capabilities
as the reference repo ahmdtaha/distributed_sigmoid_loss:

- ``DistributedSigmoidLoss`` — all-gather strategy, module-owned learnable
  ``t_prime``/``bias`` (behavioral parity with reference
  ``distributed_sigmoid_loss.py:8-48``).
- ``SigLipLoss`` — ring strategy with caller-owned ``logit_scale``/``logit_bias``
  (behavioral parity with reference ``rwightman_sigmoid_loss.py:12-124``).
- ``parallel`` — autograd-correct ring neighbour-exchange primitives and
  differentiable collectives over torch.distributed (RCCL on ROCm, gloo on CPU)
  (behavioral parity with reference ``distributed_utils.py:1-106``).
- ``ops`` — hand-written CDNA4 HIP kernels (MFMA, LDS-tiled, gfx950) fusing
  pairwise logits + temperature/bias + log-sigmoid + reduction, forward and
  backward, never materializing the full N×N logits matrix.

The compute path is PyTorch-ROCm + HIP/CDNA4 kernels + RCCL over xGMI; there
are no CUDA compatibility layers and no multi-backend dispatch.
"""

__version__ = "0.2.0"

from .losses.sigmoid_loss import DistributedSigmoidLoss, SigLipLoss
from .losses.functional import sigmoid_contrastive_loss
from .parallel.ring import (
    neighbour_exchange,
    neighbour_exchange_bidir,
    neighbour_exchange_with_grad,
    neighbour_exchange_bidir_with_grad,
)
from .parallel.collectives import all_gather_with_grad

__all__ = [
    "DistributedSigmoidLoss",
    "SigLipLoss",
    "sigmoid_contrastive_loss",
    "neighbour_exchange",
    "neighbour_exchange_bidir",
    "neighbour_exchange_with_grad",
    "neighbour_exchange_bidir_with_grad",
    "all_gather_with_grad",
]
