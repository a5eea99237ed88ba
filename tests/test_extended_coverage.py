"""Extended CPU coverage: process subgroups, wide rings, exact gradients.

Fills surfaces the reference's oracles don't reach:

- ``DistributedSigmoidLoss.forward(..., group=...)`` on a torch.distributed
  *subgroup* — including a non-trivial one ({1, 2} of a 3-rank world) whose
  group-local ranks differ from global ranks, which exercises the
  group→global peer translation in the ring's P2P exchanges (``P2POp`` peers
  are global ranks even when a group is passed).
- W=5 ring: two full bidirectional rounds and no remainder hop (the even
  hop-count case; W=3 covers one round, W=4 covers round+remainder).
- float64 ``torch.autograd.gradcheck`` of the op core — an exactness proof
  stronger than the rtol=1e-3 oracle comparisons
  (reference tolerance: ``test_distributed_sigmoid_loss.py:140-141``).
- ``torch.no_grad()`` forward through the ring strategy at W>1 (inference
  path: no saved slabs, loss value must match the grad-mode forward).
"""

import math

import pytest
import torch
import torch.distributed as dist

from distributed_sigmoid_loss_amd import DistributedSigmoidLoss
from distributed_sigmoid_loss_amd.losses.functional import (
    sigmoid_contrastive_loss,
)

from helpers import encode_shard, run_distributed


def _subgroup_step(rank, world, members, strategy):
    """Build a subgroup of ``members``; those ranks run the loss over it."""
    sub = dist.new_group(ranks=members)  # collective: every rank calls it
    if rank not in members:
        return None
    sub_world = len(members)
    sub_rank = members.index(rank)
    # Shard the (seeded, identical-everywhere) global batch by SUBGROUP rank
    # so the two strategies see the same data regardless of global rank.
    img_enc, txt_enc, zi, zt = encode_shard(sub_rank, sub_world, 4, 16)
    mod = DistributedSigmoidLoss(4, strategy=strategy)
    loss = mod(zi, zt, group=sub)
    loss.backward()
    if sub_rank == 0:
        return {
            "img": img_enc.weight.grad.clone(),
            "txt": txt_enc.weight.grad.clone(),
            "scale": mod.t_prime.grad.clone(),
            "bias": mod.bias.grad.clone(),
            "loss": loss.detach().clone(),
        }


@pytest.mark.parametrize("members", [[0, 1], [1, 2]],
                         ids=["trivial01", "offset12"])
def test_subgroup_ring_matches_all_gather(members):
    """Ring == all-gather raw grads when the loss runs over a SUBGROUP of
    the world.  ``[1, 2]`` is the discriminating case: its group-local ranks
    (0, 1) differ from its global ranks, so ring P2P peers must be
    translated group→global (collectives handle groups natively)."""
    ring = run_distributed(_subgroup_step, 3, members, "ring")
    gather = run_distributed(_subgroup_step, 3, members, "all_gather")
    lead = members[0]
    for key in ("img", "txt", "scale", "bias", "loss"):
        assert torch.allclose(ring[lead][key], gather[lead][key],
                              rtol=1e-3, atol=1e-6), key


def _ddp_step_raw(rank, world, strategy, bidir):
    img_enc, txt_enc, zi, zt = encode_shard(rank, world, 3, 8)
    mod = DistributedSigmoidLoss(3, strategy=strategy, bidir=bidir)
    loss = mod(zi, zt)
    loss.backward()
    if rank == 0:
        return {
            "img": img_enc.weight.grad.clone(),
            "txt": txt_enc.weight.grad.clone(),
            "scale": mod.t_prime.grad.clone(),
            "bias": mod.bias.grad.clone(),
            "loss": loss.detach().clone(),
        }


def test_w5_bidir_ring_even_hops():
    """W=5: divmod(4, 2) = (2, 0) — two full bidirectional rounds, NO
    remainder hop (W=3/W=4 elsewhere cover one round and round+remainder).
    Must equal the all-gather strategy's raw grads."""
    ring = run_distributed(_ddp_step_raw, 5, "ring", True)[0]
    gather = run_distributed(_ddp_step_raw, 5, "all_gather", True)[0]
    for key in ("img", "txt", "scale", "bias", "loss"):
        assert torch.allclose(ring[key], gather[key],
                              rtol=1e-3, atol=1e-6), key


def test_w8_bidir_ring_full_node_shape():
    """W=8 — the 8-GPU node shape the driver's SCALE run uses:
    divmod(7, 2) = (3, 1), i.e. three bidirectional rounds PLUS a
    remainder hop (no smaller W hits multiple rounds AND a remainder).
    Ring raw grads must equal all-gather raw grads."""
    ring = run_distributed(_ddp_step_raw, 8, "ring", True)[0]
    gather = run_distributed(_ddp_step_raw, 8, "all_gather", True)[0]
    for key in ("img", "txt", "scale", "bias", "loss"):
        assert torch.allclose(ring[key], gather[key],
                              rtol=1e-3, atol=1e-6), key


@pytest.mark.parametrize("diag_offset", [None, 0, 1])
@pytest.mark.parametrize("col_chunk", [None, 2])
def test_gradcheck_float64(diag_offset, col_chunk):
    """Exact-gradient proof of the op core (impl='torch' is the CPU
    reference the HIP kernels are numerics-tested against): float64
    gradcheck over all four differentiable inputs."""
    torch.manual_seed(7)
    zi = torch.randn(3, 4, dtype=torch.float64, requires_grad=True)
    zt = torch.randn(5, 4, dtype=torch.float64, requires_grad=True)
    tp = torch.tensor(math.log(3.0), dtype=torch.float64,
                      requires_grad=True)
    bias = torch.tensor(-2.0, dtype=torch.float64, requires_grad=True)

    def fn(zi_, zt_, tp_, bias_):
        return sigmoid_contrastive_loss(zi_, zt_, tp_, bias_,
                                        diag_offset=diag_offset,
                                        col_chunk=col_chunk, impl="torch")

    assert torch.autograd.gradcheck(fn, (zi, zt, tp, bias),
                                    raise_exception=True)


def _rw_step_w(rank, world, bidir):
    import torch.nn as nn
    from distributed_sigmoid_loss_amd import SigLipLoss
    img_enc, txt_enc, zi, zt = encode_shard(rank, world, 3, 8)
    scale = nn.Parameter(torch.tensor(math.log(10.0)))
    bias = nn.Parameter(torch.tensor(-10.0))
    loss = SigLipLoss(rank=rank, world_size=world, bidir=bidir)(
        zi, zt, scale, bias)
    loss.backward()
    if rank == 0:
        return {"img": img_enc.weight.grad.clone(),
                "txt": txt_enc.weight.grad.clone(),
                "scale": scale.grad.clone(), "bias": bias.grad.clone(),
                "loss": loss.detach().clone()}


@pytest.mark.parametrize("bidir", [True, False])
def test_sigliploss_chain_w8(bidir):
    """The reference-API SigLipLoss autograd exchange chain at the full
    8-rank node shape (7 reversed grad hops unidirectional; 3 bidir rounds
    + remainder when bidir) vs DistributedSigmoidLoss raw grads."""
    rw = run_distributed(_rw_step_w, 8, bidir)[0]
    ddp = run_distributed(_ddp_step_raw, 8, "all_gather", True)[0]
    for key in ("img", "txt", "scale", "bias", "loss"):
        assert torch.allclose(rw[key], ddp[key], rtol=1e-3, atol=1e-6), key


def _sync_comm_step(rank, world):
    from distributed_sigmoid_loss_amd.parallel.ring import set_sync_comm
    set_sync_comm(True)
    try:
        return _ddp_step_raw(rank, world, "ring", True)
    finally:
        set_sync_comm(False)


def test_sync_comm_fallback_unchanged():
    """SIGLIP_SYNC_COMM (the race-bisection debug mode): forcing
    every exchange to complete synchronously must not change gradients."""
    sync = run_distributed(_sync_comm_step, 3)[0]
    normal = run_distributed(_ddp_step_raw, 3, "ring", True)[0]
    for key in ("img", "txt", "scale", "bias", "loss"):
        assert torch.allclose(sync[key], normal[key],
                              rtol=1e-6, atol=1e-8), key


def _nograd_step(rank, world, strategy):
    _, _, zi, zt = encode_shard(rank, world, 4, 16)
    mod = DistributedSigmoidLoss(4, strategy=strategy)
    with torch.no_grad():
        inference = mod(zi.detach(), zt.detach())
    training = mod(zi, zt)
    if rank == 0:
        return {"inference": inference.clone(),
                "training": training.detach().clone()}


@pytest.mark.parametrize("strategy", ["ring", "all_gather"])
def test_no_grad_forward_matches(strategy):
    """Inference-mode forward at W=2 (no saved state, want_grad=False)
    returns the same loss value as the training forward."""
    out = run_distributed(_nograd_step, 2, strategy)[0]
    assert torch.allclose(out["inference"], out["training"],
                          rtol=1e-6, atol=1e-8)
