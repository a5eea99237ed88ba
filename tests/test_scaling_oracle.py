"""Scaling oracle: N-rank gradients == 1-rank gradients on the same global batch.

Re-derivation of the reference's main correctness test
(``test_distributed_sigmoid_loss.py:122-141``): the distributed loss under
manual DDP grad averaging must reproduce the single-rank loss's encoder
gradients at rtol=1e-3 — for both comm strategies.
"""

import pytest
import torch

from distributed_sigmoid_loss_amd import DistributedSigmoidLoss
from distributed_sigmoid_loss_amd.parallel import average_gradients

from helpers import encode_shard, run_distributed


def ddp_step(rank, world, batch_per_rank, emb_dim, strategy, average,
             quant="bf16"):
    img_enc, txt_enc, zi, zt = encode_shard(rank, world, batch_per_rank,
                                            emb_dim)
    bidir = strategy == "ring_bidir"
    loss_mod = DistributedSigmoidLoss(
        batch_per_rank, strategy="ring" if bidir else strategy, bidir=bidir,
        quant=quant)
    loss = loss_mod(zi, zt)
    loss.backward()
    if average:
        average_gradients(img_enc)
        average_gradients(txt_enc)
        average_gradients(loss_mod)
    if rank == 0:
        return {
            "img": img_enc.weight.grad.clone(),
            "txt": txt_enc.weight.grad.clone(),
            "t_prime": loss_mod.t_prime.grad.clone(),
            "bias": loss_mod.bias.grad.clone(),
            "loss": loss.detach().clone(),
        }


@pytest.mark.parametrize("strategy", ["all_gather", "ring"])
@pytest.mark.parametrize("world,batch", [(2, 4), (3, 3)])
@pytest.mark.parametrize("emb_dim", [8, 64])
def test_n_rank_matches_single_rank(strategy, world, batch, emb_dim):
    multi = run_distributed(ddp_step, world, batch, emb_dim, strategy, True)[0]
    single = run_distributed(ddp_step, 1, world * batch, emb_dim, strategy,
                             True)[0]
    for key in ("img", "txt", "t_prime", "bias"):
        assert torch.allclose(multi[key], single[key], rtol=1e-3, atol=1e-6), \
            f"{key} grads diverge: {multi[key]} vs {single[key]}"


@pytest.mark.parametrize("strategy", ["all_gather", "ring"])
@pytest.mark.parametrize("emb_dim", [128, 512])
def test_n_rank_matches_single_rank_wide_dims(strategy, emb_dim):
    """Reference's wide-dim sweep (test_distributed_sigmoid_loss.py:144-148
    runs d up to 512) at W=2, batch 4."""
    multi = run_distributed(ddp_step, 2, 4, emb_dim, strategy, True)[0]
    single = run_distributed(ddp_step, 1, 8, emb_dim, strategy, True)[0]
    for key in ("img", "txt", "t_prime", "bias"):
        assert torch.allclose(multi[key], single[key], rtol=1e-3, atol=1e-6), \
            f"{key} grads diverge at d={emb_dim}"


@pytest.mark.parametrize("world", [2, 3])
def test_ring_matches_all_gather_raw_grads(world):
    """Both strategies of DistributedSigmoidLoss must be numerically
    interchangeable at the raw (un-averaged) gradient level."""
    ring = run_distributed(ddp_step, world, 4, 16, "ring", False)[0]
    gather = run_distributed(ddp_step, world, 4, 16, "all_gather", False)[0]
    for key in ("img", "txt", "t_prime", "bias", "loss"):
        assert torch.allclose(ring[key], gather[key], rtol=1e-4, atol=1e-7), key


@pytest.mark.parametrize("world", [3, 4])
def test_bidir_ring_matches_all_gather_raw_grads(world):
    """The hop-halved bidirectional ring (W=3: one bidir round; W=4: bidir
    round + unidirectional remainder hop) is numerically interchangeable
    with the all-gather strategy."""
    ring = run_distributed(ddp_step, world, 4, 16, "ring_bidir", False)[0]
    gather = run_distributed(ddp_step, world, 4, 16, "all_gather", False)[0]
    for key in ("img", "txt", "t_prime", "bias", "loss"):
        assert torch.allclose(ring[key], gather[key], rtol=1e-4, atol=1e-7), key


def test_bidir_ring_w5_rotation():
    """W=5: TWO bidirectional rounds, exercising the re-post of received
    chunks (the rotation ``to_left, to_right ← received pair``) that W≤4
    never reaches."""
    ring = run_distributed(ddp_step, 5, 2, 16, "ring_bidir", False)[0]
    gather = run_distributed(ddp_step, 5, 2, 16, "all_gather", False)[0]
    for key in ("img", "txt", "t_prime", "bias", "loss"):
        assert torch.allclose(ring[key], gather[key], rtol=1e-4, atol=1e-7), key


@pytest.mark.parametrize("world,batch", [(3, 3), (4, 2)])
def test_bidir_ring_scaling_oracle(world, batch):
    """N-rank bidir-ring grads == 1-rank grads (the reference's main
    correctness property, test_distributed_sigmoid_loss.py:122-141)."""
    multi = run_distributed(ddp_step, world, batch, 16, "ring_bidir", True)[0]
    single = run_distributed(ddp_step, 1, world * batch, 16, "all_gather",
                             True)[0]
    for key in ("img", "txt", "t_prime", "bias"):
        assert torch.allclose(multi[key], single[key], rtol=1e-3, atol=1e-6), key


@pytest.mark.parametrize("world,strategy",
                         [(3, "ring"), (4, "ring_bidir"),
                          (8, "ring_bidir")])
def test_fp8_wire_ring_cpu_routing(world, strategy):
    """fp8 ring ships e4m3 + scale over the wire; on CPU the received
    chunks are dequantized for the torch path.  Validates the wire
    routing/scale pairing (incl. bidir forwarding) against the all_gather
    strategy — which on CPU computes on unquantized values — so the
    tolerance is the e4m3 quantization level, not exactness."""
    ring = run_distributed(ddp_step, world, 4, 32, strategy, False, "fp8")[0]
    gather = run_distributed(ddp_step, world, 4, 32, "all_gather", False,
                             "bf16")[0]
    for key in ("img", "txt", "loss"):
        a, b = ring[key].float(), gather[key].float()
        rel = (a - b).norm() / b.norm().clamp(min=1e-12)
        assert rel < 8e-2, (key, rel)


def test_bidir_ring_w8_scale_shape():
    """W=8 — the exact control-flow shape of the driver's 8-GPU SCALE run:
    THREE bidirectional rounds (two chunk re-posts) plus the unidirectional
    remainder hop."""
    ring = run_distributed(ddp_step, 8, 2, 16, "ring_bidir", False)[0]
    gather = run_distributed(ddp_step, 8, 2, 16, "all_gather", False)[0]
    for key in ("img", "txt", "t_prime", "bias", "loss"):
        assert torch.allclose(ring[key], gather[key], rtol=1e-4, atol=1e-7), key
