"""Module-level contracts: state_dict serialization of the loss parameters
(reference README.md:20 — users hand t_prime/bias to the optimizer; the only
persistent state, serialized via the standard state_dict path), construction
validation, and single-process forward behavior."""

import math

import pytest
import torch
import torch.nn.functional as F

from distributed_sigmoid_loss_amd import DistributedSigmoidLoss, SigLipLoss


def test_state_dict_roundtrip():
    mod = DistributedSigmoidLoss(8)
    assert math.isclose(float(mod.t_prime.detach()), math.log(10.0), rel_tol=1e-6)
    assert float(mod.bias.detach()) == -10.0
    with torch.no_grad():
        mod.t_prime.fill_(1.5)
        mod.bias.fill_(-3.25)
    sd = mod.state_dict()
    assert set(sd.keys()) == {"t_prime", "bias"}

    mod2 = DistributedSigmoidLoss(8)
    mod2.load_state_dict(sd)
    assert float(mod2.t_prime.detach()) == 1.5
    assert float(mod2.bias.detach()) == -3.25


def test_loss_params_in_optimizer():
    mod = DistributedSigmoidLoss(4)
    opt = torch.optim.SGD(mod.parameters(), lr=0.1)
    zi = F.normalize(torch.randn(4, 16), dim=-1)
    zt = F.normalize(torch.randn(4, 16), dim=-1)
    before = (float(mod.t_prime.detach()), float(mod.bias.detach()))
    mod(zi, zt).backward()
    opt.step()
    after = (float(mod.t_prime.detach()), float(mod.bias.detach()))
    assert before != after  # both params updated through the optimizer


def test_invalid_args():
    with pytest.raises(ValueError):
        DistributedSigmoidLoss(4, strategy="bogus")
    with pytest.raises(ValueError):
        DistributedSigmoidLoss(4, quant="int4")


def test_single_process_no_dist():
    """Works without an initialized process group (world_size=1 semantics) —
    the reference requires dist.init (distributed_sigmoid_loss.py:37)."""
    mod = DistributedSigmoidLoss(6)
    zi = F.normalize(torch.randn(6, 32), dim=-1)
    zt = F.normalize(torch.randn(6, 32), dim=-1)
    loss = mod(zi, zt)
    assert loss.ndim == 0 and torch.isfinite(loss)

    sig = SigLipLoss(rank=0, world_size=1)
    scale = torch.tensor(math.log(10.0))
    bias = torch.tensor(-10.0)
    loss2 = sig(zi, zt, scale, bias)
    # DistributedSigmoidLoss normalizes once by b; SigLipLoss per chunk by b.
    assert torch.allclose(loss, loss2, rtol=1e-5)
