"""Unit tests for the ring neighbour-exchange primitives (the reference has
no unit tests of these — SURVEY §4 calls for adding them)."""

import pytest
import torch

from distributed_sigmoid_loss_amd.parallel import (
    neighbour_exchange,
    neighbour_exchange_bidir,
    neighbour_exchange_with_grad,
    neighbour_exchange_bidir_with_grad,
    neighbour_exchange_start,
)

from helpers import run_distributed


def _exchange_roundtrip(rank, world):
    x = torch.full((3, 4), float(rank))
    left = (rank - 1 + world) % world
    right = (rank + 1) % world
    recv = neighbour_exchange(left, right, x)
    assert torch.all(recv == float(left)), (rank, recv)
    return True


def _exchange_bidir(rank, world):
    left = (rank - 1 + world) % world
    right = (rank + 1) % world
    to_left = torch.full((2, 2), float(rank * 10))
    to_right = torch.full((2, 2), float(rank * 10 + 1))
    from_right, from_left = neighbour_exchange_bidir(left, right, to_left,
                                                     to_right)
    # Right neighbour sent its to_left; left neighbour sent its to_right.
    assert torch.all(from_right == float(right * 10)), (rank, from_right)
    assert torch.all(from_left == float(left * 10 + 1)), (rank, from_left)
    return True


def _exchange_nonblocking(rank, world):
    left = (rank - 1 + world) % world
    right = (rank + 1) % world
    h = neighbour_exchange_start(left, right, torch.full((5,), float(rank)))
    recv = h.wait()[0]
    assert torch.all(recv == float(left))
    h.wait()  # idempotent
    return True


def _exchange_grad_routing(rank, world):
    """grad of the sent tensor must come back from the receiving rank:
    rank r sends x_r to the right; the right neighbour scales the received
    tensor by (its rank + 1); so dL/dx_r == right_rank + 1."""
    left = (rank - 1 + world) % world
    right = (rank + 1) % world
    x = torch.ones(3, requires_grad=True)
    recv = neighbour_exchange_with_grad(left, right, x)
    loss = (recv * float(rank + 1)).sum()
    loss.backward()
    assert torch.allclose(x.grad, torch.full((3,), float(right + 1))), \
        (rank, x.grad)
    return True


def _exchange_bidir_grad_routing(rank, world):
    left = (rank - 1 + world) % world
    right = (rank + 1) % world
    x_to_left = torch.ones(3, requires_grad=True)
    x_to_right = torch.ones(3, requires_grad=True)
    from_right, from_left = neighbour_exchange_bidir_with_grad(
        left, right, x_to_left, x_to_right)
    # from_right gets weight (rank+1)*10; from_left gets weight (rank+1).
    loss = (from_right * float((rank + 1) * 10)).sum() \
        + (from_left * float(rank + 1)).sum()
    loss.backward()
    # x_to_left went to the left neighbour and landed in its from_right term.
    assert torch.allclose(x_to_left.grad,
                          torch.full((3,), float((left + 1) * 10))), \
        (rank, x_to_left.grad)
    # x_to_right went to the right neighbour's from_left term.
    assert torch.allclose(x_to_right.grad,
                          torch.full((3,), float(right + 1))), \
        (rank, x_to_right.grad)
    return True


@pytest.mark.parametrize("world", [2, 3])
def test_exchange_roundtrip(world):
    run_distributed(_exchange_roundtrip, world)


@pytest.mark.parametrize("world", [3, 4])
def test_exchange_bidir(world):
    run_distributed(_exchange_bidir, world)


def test_exchange_nonblocking():
    run_distributed(_exchange_nonblocking, 3)


@pytest.mark.parametrize("world", [2, 3])
def test_exchange_grad_routing(world):
    run_distributed(_exchange_grad_routing, world)


def test_exchange_bidir_grad_routing():
    run_distributed(_exchange_bidir_grad_routing, 3)


def _quantized_roundtrip(rank, world):
    from distributed_sigmoid_loss_amd.parallel.ring import (
        quantized_exchange_start,
    )
    left = (rank - 1 + world) % world
    right = (rank + 1) % world
    x = torch.randn(8, 16) * (rank + 1)
    from distributed_sigmoid_loss_amd import ops
    q, s = ops._quant_fp8(x)
    h = quantized_exchange_start(left, right, q, s.reshape(1).float())
    rq_u8, rs = h.wait()
    rq = rq_u8.view(torch.float8_e4m3fn)
    # verify by echoing: exchange back and compare bitwise with what we sent
    h2 = quantized_exchange_start(right, left, rq, rs)
    back_q, back_s = h2.wait()
    if world == 2:
        # two hops around a 2-ring returns our own payload
        assert torch.equal(back_q, q.view(torch.uint8))
        assert torch.equal(back_s, s.reshape(1).float())
    assert rq.shape == q.shape and float(rs[0]) > 0
    return True


def test_quantized_exchange_roundtrip():
    run_distributed(_quantized_roundtrip, 2)
