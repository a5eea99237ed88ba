#!/usr/bin/env python3
"""End-to-end distributed SigLIP contrastive training example.

Trains a two-tower model with the fused MI355X loss kernels over RCCL/xGMI.
Synthetic data (no network access in this environment); swap `synthetic_batch`
for a real image/text dataloader in production.

Single GPU:
    python examples/train_siglip.py --steps 50

8 GPUs (one process per GPU over RCCL):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/train_siglip.py --steps 50

CPU smoke (gloo):
    python examples/train_siglip.py --device cpu --batch-per-gpu 8 --dim 64
"""

from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from distributed_sigmoid_loss_amd import DistributedSigmoidLoss
from distributed_sigmoid_loss_amd.models import TwoTowerModel
from distributed_sigmoid_loss_amd.parallel import average_gradients
from distributed_sigmoid_loss_amd.utils import init_from_env, set_seed
from distributed_sigmoid_loss_amd.utils.profiling import PhaseTimer


def synthetic_batch(b, dim, device, dtype, seed):
    # Device-side generation — a host dataloader would overlap H2D copies
    # with compute via a prefetching pipeline; synthetic data skips that.
    # Seeded per (step, rank) so each rank's shard is distinct — with a
    # shared generator every rank would draw identical batches and the
    # cross-rank negatives would duplicate the positives.
    g = torch.Generator(device=device).manual_seed(seed)
    img = torch.randn(b, dim, device=device, dtype=dtype, generator=g)
    txt = torch.randn(b, dim, device=device, dtype=dtype, generator=g)
    return img, txt


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--batch-per-gpu", type=int, default=4096)
    p.add_argument("--dim", type=int, default=768)
    p.add_argument("--lr", type=float, default=1e-3)
    p.add_argument("--strategy", choices=["ring", "all_gather"],
                   default="ring")
    p.add_argument("--quant", choices=["bf16", "fp8", "mixed"], default="bf16")
    p.add_argument("--device", default=None)
    p.add_argument("--log-every", type=int, default=10)
    args = p.parse_args()

    rank, local_rank, world = init_from_env()
    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if device == "cuda" else torch.float32
    set_seed(1234)  # identical tower init on every rank

    model = TwoTowerModel(args.dim, args.dim).to(device=device, dtype=dtype)
    loss_mod = DistributedSigmoidLoss(
        args.batch_per_gpu, strategy=args.strategy,
        quant=args.quant if device == "cuda" else "bf16").to(device)
    # Loss params (t_prime, bias) must reach the optimizer too
    # (reference README.md:20).
    opt = torch.optim.AdamW(
        list(model.parameters()) + list(loss_mod.parameters()), lr=args.lr)

    timer = PhaseTimer(enabled=True, use_cuda=(device == "cuda"))
    t0 = time.perf_counter()
    for step in range(args.steps):
        img, txt = synthetic_batch(args.batch_per_gpu, args.dim, device,
                                   dtype, step * world + rank)
        opt.zero_grad(set_to_none=True)
        with timer.phase("encode"):
            zi, zt = model(img, txt)
        with timer.phase("loss"):
            loss = loss_mod(zi, zt)
        with timer.phase("backward"):
            loss.backward()
        with timer.phase("grad_avg"):
            if world > 1:
                average_gradients(model)
                average_gradients(loss_mod)
        with timer.phase("opt"):
            opt.step()
        timer.step_end()

        if rank == 0 and (step + 1) % args.log_every == 0:
            phases = "  ".join(f"{k}={v:.2f}ms"
                               for k, v in timer.summary().items())
            print(f"step {step + 1:4d}  loss={float(loss.detach()):.4f}  "
                  f"t={float(loss_mod.t_prime.detach().exp()):.3f}  "
                  f"bias={float(loss_mod.bias.detach()):.3f}  {phases}", flush=True)
            timer.rows.clear()

    if device == "cuda":
        torch.cuda.synchronize()
    if rank == 0:
        total = time.perf_counter() - t0
        pairs = args.batch_per_gpu * world * args.steps
        print(f"done: {args.steps} steps, {pairs / total:,.0f} pairs/s "
              f"aggregate over {world} rank(s)")
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
