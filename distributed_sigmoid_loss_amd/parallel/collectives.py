"""Differentiable collectives for the all-gather loss strategy.

The reference uses ``torch.distributed.nn.functional.all_gather``
(``distributed_sigmoid_loss.py:5,35``), whose backward is a reduce-scatter on
NCCL-family backends.  Here we implement the pair natively, sized for RCCL over
xGMI:

- forward: one ``all_gather_into_tensor`` into a single flat ``(W·b, d)``
  buffer (one RCCL ring all-gather — per-link bound ≈153 GB/s on xGMI — instead
  of W list entries);
- backward: one ``reduce_scatter_tensor(SUM)`` so each rank ends with the summed
  gradient for its own shard.  On gloo (CPU CI), which lacks reduce-scatter, we
  fall back to all_reduce + slice — same semantics.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


def _backend_is_gloo(group) -> bool:
    try:
        return dist.get_backend(group) == "gloo"
    except Exception:  # pragma: no cover - defensive
        return True


class _AllGatherFlat(torch.autograd.Function):
    @staticmethod
    def forward(ctx, group, tensor):
        ctx.group = group
        world = dist.get_world_size(group)
        ctx.world = world
        ctx.rank = dist.get_rank(group)
        tensor = tensor.contiguous()
        out = torch.empty((world * tensor.shape[0],) + tuple(tensor.shape[1:]),
                          dtype=tensor.dtype, device=tensor.device)
        dist.all_gather_into_tensor(out, tensor, group=group)
        return out

    @staticmethod
    def backward(ctx, grad_output):
        grad_output = grad_output.contiguous()
        b = grad_output.shape[0] // ctx.world
        if _backend_is_gloo(ctx.group):
            # gloo has no reduce_scatter_tensor: all-reduce then slice.
            dist.all_reduce(grad_output, op=dist.ReduceOp.SUM, group=ctx.group)
            grad_local = grad_output[ctx.rank * b:(ctx.rank + 1) * b].clone()
        else:
            grad_local = torch.empty_like(grad_output[:b])
            dist.reduce_scatter_tensor(grad_local, grad_output,
                                       op=dist.ReduceOp.SUM, group=ctx.group)
        return None, grad_local


def all_gather_with_grad(tensor: torch.Tensor, group=None) -> torch.Tensor:
    """Differentiable all-gather returning one flat ``(W·b, …)`` tensor.

    Gradient semantics match ``torch.distributed.nn.functional.all_gather``
    (used at reference ``distributed_sigmoid_loss.py:35``): each rank's input
    gradient is the sum over ranks of the gradient of its own shard.
    """
    if not dist.is_available() or not dist.is_initialized() \
            or dist.get_world_size(group) == 1:
        return tensor
    return _AllGatherFlat.apply(group, tensor)


def average_gradients(module: torch.nn.Module, group=None) -> None:
    """Manual DDP gradient averaging: all_reduce(SUM) then divide by W.

    Parity with the reference test harness
    (``test_distributed_sigmoid_loss.py:79-83``).  For real training prefer
    ``torch.nn.parallel.DistributedDataParallel`` with bucketed RCCL
    all-reduce overlapped with backward; this helper exists for the
    equivalence oracles and small parameter sets (e.g. the loss's
    ``t_prime``/``bias``).
    """
    if not dist.is_available() or not dist.is_initialized():
        return
    world = dist.get_world_size(group)
    if world == 1:
        return
    for p in module.parameters():
        if p.grad is not None:
            dist.all_reduce(p.grad, op=dist.ReduceOp.SUM, group=group)
            p.grad.div_(world)
