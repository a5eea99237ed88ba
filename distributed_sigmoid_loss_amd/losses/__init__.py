from .sigmoid_loss import DistributedSigmoidLoss, SigLipLoss
from .functional import sigmoid_contrastive_loss

__all__ = ["DistributedSigmoidLoss", "SigLipLoss", "sigmoid_contrastive_loss"]
