"""Autograd-correct ring neighbour-exchange primitives over torch.distributed.

Behavioral parity with the reference's L1 layer (``distributed_utils.py:10-106``):
``neighbour_exchange`` / ``neighbour_exchange_bidir`` post paired isend/irecv via
``P2POp`` + ``batch_isend_irecv``; the ``*_with_grad`` autograd wrappers re-send
``grad_output`` along the reversed route in backward, so gradients hop backwards
around the ring one exchange per backward step.

MI355X mapping: with ``init_process_group("nccl")`` on ROCm these P2P ops are
RCCL send/recv over xGMI point-to-point links (7 links × ≈153 GB/s per GPU on an
8-GPU node).  A unidirectional hop uses one link per direction; the bidirectional
variant engages two links simultaneously and halves the hop count to
``⌈(W-1)/2⌉``.  RCCL runs these on its own internal HIP stream, so an exchange
issued *before* the compute on the previously-received chunk overlaps with it —
see ``RingHandle`` and ``losses/sigmoid_loss.py::SigLipLoss`` for the
double-buffered schedule (the reference serializes comm and compute by calling
``req.wait()`` immediately, ``distributed_utils.py:25-26``).
"""

from __future__ import annotations

import os
from typing import Sequence

import torch
import torch.distributed as dist

# Debug escape hatch (SURVEY §5 race-detection plan): SIGLIP_SYNC_COMM=1
# makes every exchange complete before returning, serializing comm against
# compute — use to bisect overlap bugs (with HIP_LAUNCH_BLOCKING=1 for
# kernel-side ordering).
_SYNC_COMM = os.environ.get("SIGLIP_SYNC_COMM", "0") == "1"


def set_sync_comm(enabled: bool) -> None:
    global _SYNC_COMM
    _SYNC_COMM = enabled


class RingHandle:
    """In-flight neighbour exchange: receive buffers plus pending requests.

    Lets callers overlap the wire time with compute: start the exchange, do
    compute on the previous chunk, then ``wait()`` for the received tensors.
    """

    __slots__ = ("_reqs", "_recv", "_done")

    def __init__(self, reqs: Sequence, recv: Sequence[torch.Tensor]):
        self._reqs = reqs
        self._recv = recv
        self._done = False

    def wait(self):
        if not self._done:
            for r in self._reqs:
                r.wait()
            self._done = True
        return self._recv


def isend_irecv(sends, send_ranks, recvs, recv_ranks, group=None) -> RingHandle:
    """Post a batched set of isend/irecv pairs and return a waitable handle.

    ``sends[i]`` goes to ``send_ranks[i]``; ``recvs[i]`` is filled from
    ``recv_ranks[i]``.  All buffers must be contiguous.
    """
    ops = []
    for t, r in zip(sends, send_ranks):
        ops.append(dist.P2POp(dist.isend, t.contiguous(), r, group=group))
    for t, r in zip(recvs, recv_ranks):
        ops.append(dist.P2POp(dist.irecv, t, r, group=group))
    reqs = dist.batch_isend_irecv(ops)
    handle = RingHandle(reqs, list(recvs))
    if _SYNC_COMM:
        handle.wait()
    return handle


def neighbour_exchange(from_rank: int, to_rank: int, tensor: torch.Tensor,
                       group=None) -> torch.Tensor:
    """Blocking one-hop ring exchange: send to ``to_rank``, receive from
    ``from_rank``.  Parity: reference ``distributed_utils.py:10-27``."""
    recv = torch.empty_like(tensor)
    h = isend_irecv([tensor], [to_rank], [recv], [from_rank], group=group)
    return h.wait()[0]


def neighbour_exchange_start(from_rank: int, to_rank: int, tensor: torch.Tensor,
                             group=None) -> RingHandle:
    """Non-blocking variant of :func:`neighbour_exchange` for comm/compute
    overlap; call ``.wait()`` on the returned handle before reading."""
    recv = torch.empty_like(tensor)
    return isend_irecv([tensor], [to_rank], [recv], [from_rank], group=group)


def neighbour_exchange_bidir(left_rank: int, right_rank: int,
                             tensor_to_left: torch.Tensor,
                             tensor_to_right: torch.Tensor,
                             group=None):
    """Blocking bidirectional exchange (two simultaneous ring hops).

    Returns ``(tensor_from_right, tensor_from_left)`` — same contract as the
    reference (``distributed_utils.py:30-62``).  On an xGMI-connected node the
    two directions ride two distinct point-to-point links.
    """
    h = neighbour_exchange_bidir_start(left_rank, right_rank, tensor_to_left,
                                       tensor_to_right, group=group)
    out = h.wait()
    return out[0], out[1]


def neighbour_exchange_bidir_start(left_rank: int, right_rank: int,
                                   tensor_to_left: torch.Tensor,
                                   tensor_to_right: torch.Tensor,
                                   group=None) -> RingHandle:
    recv_from_right = torch.empty_like(tensor_to_left)
    recv_from_left = torch.empty_like(tensor_to_right)
    return isend_irecv(
        [tensor_to_right, tensor_to_left], [right_rank, left_rank],
        [recv_from_right, recv_from_left], [right_rank, left_rank],
        group=group)


def quantized_exchange_start(from_rank: int, to_rank: int,
                             q: torch.Tensor, s: torch.Tensor,
                             group=None) -> RingHandle:
    """One-hop exchange of a QUANTIZED chunk: e4m3 payload (sent as its
    uint8 view — collectives do not take float8 dtypes) plus its fp32
    scale.  Halves the xGMI bytes per hop versus shipping bf16; the
    receiver's dequantization is bitwise the sender's.  ``wait()`` returns
    ``[recv_q_u8, recv_s]``."""
    qa = q.view(torch.uint8).contiguous()
    recv_q = torch.empty_like(qa)
    recv_s = torch.empty_like(s)
    return isend_irecv([qa, s], [to_rank, to_rank],
                       [recv_q, recv_s], [from_rank, from_rank], group=group)


def quantized_exchange_bidir_start(left_rank: int, right_rank: int,
                                   q_left: torch.Tensor, s_left: torch.Tensor,
                                   q_right: torch.Tensor,
                                   s_right: torch.Tensor,
                                   group=None) -> RingHandle:
    """Bidirectional quantized exchange (two xGMI links).  ``wait()``
    returns ``[q_from_right, s_from_right, q_from_left, s_from_left]``."""
    ql = q_left.view(torch.uint8).contiguous()
    qr = q_right.view(torch.uint8).contiguous()
    r_qr = torch.empty_like(ql)
    r_sr = torch.empty_like(s_left)
    r_ql = torch.empty_like(qr)
    r_sl = torch.empty_like(s_right)
    return isend_irecv(
        [qr, s_right, ql, s_left],
        [right_rank, right_rank, left_rank, left_rank],
        [r_qr, r_sr, r_ql, r_sl],
        [right_rank, right_rank, left_rank, left_rank], group=group)


class NeighbourExchange(torch.autograd.Function):
    """Differentiable one-hop exchange; backward performs the mirror-image
    exchange of ``grad_output`` (reference ``distributed_utils.py:65-77``)."""

    @staticmethod
    def forward(ctx, from_rank, to_rank, group, tensor):
        ctx.group = group
        ctx.from_rank = from_rank
        ctx.to_rank = to_rank
        return neighbour_exchange(from_rank, to_rank, tensor, group=group)

    @staticmethod
    def backward(ctx, grad_output):
        return (None, None, None) + (
            NeighbourExchange.apply(ctx.to_rank, ctx.from_rank, ctx.group,
                                    grad_output),
        )


def neighbour_exchange_with_grad(from_rank, to_rank, tensor, group=None):
    return NeighbourExchange.apply(from_rank, to_rank, group, tensor)


class NeighbourExchangeBidir(torch.autograd.Function):
    """Differentiable bidirectional exchange; backward swaps left/right
    (reference ``distributed_utils.py:84-98``)."""

    @staticmethod
    def forward(ctx, left_rank, right_rank, group, tensor_to_left,
                tensor_to_right):
        ctx.group = group
        ctx.left_rank = left_rank
        ctx.right_rank = right_rank
        return neighbour_exchange_bidir(left_rank, right_rank, tensor_to_left,
                                        tensor_to_right, group=group)

    @staticmethod
    def backward(ctx, *grad_outputs):
        return (None, None, None) + NeighbourExchangeBidir.apply(
            ctx.right_rank, ctx.left_rank, ctx.group, *grad_outputs)


def neighbour_exchange_bidir_with_grad(left_rank, right_rank, tensor_to_left,
                                       tensor_to_right, group=None):
    return NeighbourExchangeBidir.apply(left_rank, right_rank, group,
                                        tensor_to_left, tensor_to_right)
