"""Property-based tests of the CPU reference op (hypothesis).

The CPU ``impl='torch'`` path is the oracle every GPU kernel numerics test
compares against, so it gets its own independent check: against a fully
naive materialized-labels implementation (the reference's literal algebra,
``distributed_sigmoid_loss.py:22-33``), across random shapes, diagonal
offsets (including out-of-range ones) and column chunkings.
"""

import math

import pytest
import torch
import torch.nn.functional as F
from hypothesis import given, settings, strategies as st

from distributed_sigmoid_loss_amd.losses.functional import (
    sigmoid_contrastive_loss,
    chunk_loss_bwd,
)


def naive_loss(zi, zt, tp, bias, diag):
    """The reference's literal algebra with materialized labels."""
    logits = zi @ zt.T * tp.exp() + bias
    lab = -torch.ones_like(logits)
    if diag is not None:
        for i in range(zi.shape[0]):
            j = i + diag
            if 0 <= j < zt.shape[0]:
                lab[i, j] = 1.0
    return -F.logsigmoid(lab * logits).sum()


shapes = st.tuples(st.integers(1, 9), st.integers(1, 11), st.integers(1, 6))


@settings(max_examples=60, deadline=None)
@given(shape=shapes,
       diag=st.one_of(st.none(), st.integers(-12, 12)),
       col_chunk=st.one_of(st.none(), st.integers(1, 8)),
       seed=st.integers(0, 2 ** 16))
def test_loss_matches_naive_property(shape, diag, col_chunk, seed):
    b, n, d = shape
    g = torch.Generator().manual_seed(seed)
    zi = F.normalize(torch.randn(b, d, generator=g, dtype=torch.float64),
                     dim=-1)
    zt = F.normalize(torch.randn(n, d, generator=g, dtype=torch.float64),
                     dim=-1)
    tp = torch.tensor(math.log(10.0), dtype=torch.float64)
    bias = torch.tensor(-10.0, dtype=torch.float64)
    got = sigmoid_contrastive_loss(zi, zt, tp, bias, diag_offset=diag,
                                   col_chunk=col_chunk, impl="torch")
    want = naive_loss(zi, zt, tp, bias, diag)
    assert torch.allclose(got, want, rtol=1e-10, atol=1e-10)


@settings(max_examples=40, deadline=None)
@given(shape=shapes,
       diag=st.one_of(st.none(), st.integers(-3, 12)),
       col_chunk=st.one_of(st.none(), st.integers(1, 8)),
       go=st.floats(0.1, 3.0),
       seed=st.integers(0, 2 ** 16))
def test_manual_bwd_matches_autograd_property(shape, diag, col_chunk, go,
                                              seed):
    """The hand-written chunked backward (what the distributed autograd
    Functions call) == autograd of the forward, for every random config."""
    b, n, d = shape
    g = torch.Generator().manual_seed(seed)
    zi = torch.randn(b, d, generator=g, dtype=torch.float64,
                     requires_grad=True)
    zt = torch.randn(n, d, generator=g, dtype=torch.float64,
                     requires_grad=True)
    tp = torch.tensor(0.7, dtype=torch.float64, requires_grad=True)
    bias = torch.tensor(-1.3, dtype=torch.float64, requires_grad=True)
    loss = sigmoid_contrastive_loss(zi, zt, tp, bias, diag_offset=diag,
                                    impl="torch")
    go_t = torch.tensor(go, dtype=torch.float64)
    loss.backward(go_t)
    dzi, dzt, dtp, dbias = chunk_loss_bwd(
        zi.detach(), zt.detach(), tp.detach(), bias.detach(), diag, go_t,
        col_chunk=col_chunk, impl="torch")
    assert torch.allclose(dzi, zi.grad, rtol=1e-9, atol=1e-9)
    assert torch.allclose(dzt, zt.grad, rtol=1e-9, atol=1e-9)
    assert torch.allclose(dtp, tp.grad, rtol=1e-9, atol=1e-9)
    assert torch.allclose(dbias, bias.grad, rtol=1e-9, atol=1e-9)
