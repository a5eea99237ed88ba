#!/usr/bin/env python3
"""Decompose the training-step backward: loss-op backward vs tower backward
vs elementwise passes.  Run on a GPU box."""

from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from distributed_sigmoid_loss_amd import ops, DistributedSigmoidLoss
from distributed_sigmoid_loss_amd.models import TwoTowerModel


def time_fn(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def main():
    b, d = 32768, 768
    dev = "cuda"
    torch.manual_seed(0)
    quant = sys.argv[1] if len(sys.argv) > 1 else "bf16"

    model = TwoTowerModel(d, d).to(device=dev, dtype=torch.bfloat16)
    loss_mod = DistributedSigmoidLoss(b, quant=quant).cuda()
    img = torch.randn(b, d, device=dev, dtype=torch.bfloat16)
    txt = torch.randn(b, d, device=dev, dtype=torch.bfloat16)

    # 1. encode fwd
    t_enc = time_fn(lambda: model(img, txt))
    print(f"encode fwd            : {t_enc:7.3f} ms")

    # 2. loss fwd only (on detached embeddings that require grad)
    zi0, zt0 = model(img, txt)
    zi = zi0.detach().requires_grad_(True)
    zt = zt0.detach().requires_grad_(True)
    t_lf = time_fn(lambda: loss_mod(zi, zt))
    print(f"loss fwd (fwd+g)      : {t_lf:7.3f} ms")

    # 3. loss fwd + loss bwd (grads wrt embeddings + params only)
    def loss_fb():
        l = loss_mod(zi, zt)
        torch.autograd.grad(l, [zi, zt, loss_mod.t_prime, loss_mod.bias])
    t_lfb = time_fn(loss_fb)
    print(f"loss fwd+bwd          : {t_lfb:7.3f} ms   (bwd ≈ {t_lfb - t_lf:5.3f})")

    # 4. full fwd+bwd through towers
    def full_fb():
        for p in model.parameters():
            p.grad = None
        a, t_ = model(img, txt)
        loss_mod(a, t_).backward()
    t_full = time_fn(full_fb)
    print(f"full fwd+bwd          : {t_full:7.3f} ms   "
          f"(towers fwd+bwd+norm ≈ {t_full - t_lfb:5.3f})")

    # 5. components of bwd_from_g
    if quant == "bf16":
        _, g, _ = ops.siglip_fwd_g(zi.detach(), zt.detach(),
                                   loss_mod.t_prime.detach(),
                                   loss_mod.bias.detach(), 0)
        sc = torch.tensor(2.3, device=dev)
        t_a = time_fn(lambda: (g.T @ zi.detach()))
        t_a2 = time_fn(lambda: ((g.T @ zi.detach()) * sc).to(torch.bfloat16))
        t_b = time_fn(lambda: (g @ zt.detach()))
        t_b2 = time_fn(lambda: ((g @ zt.detach()) * sc).to(torch.bfloat16))
        print(f"gT@zi raw / +scale+cast: {t_a:6.3f} / {t_a2:6.3f} ms")
        print(f"g@zt  raw / +scale+cast: {t_b:6.3f} / {t_b2:6.3f} ms")

    # 5b. fp8 backward pieces
    if quant == "fp8":
        zi_d, zt_d = zi.detach(), zt.detach()
        qc = ops.quantize_fp8_pair(zi_d, zt_d)
        t_q = time_fn(lambda: ops.quantize_fp8_pair(zi_d, zt_d))
        print(f"quantize pair         : {t_q:7.3f} ms")
        buf, g8, gt8 = ops.siglip_fwd_g(zi_d, zt_d,
                                        loss_mod.t_prime.detach(),
                                        loss_mod.bias.detach(), 0,
                                        quant="fp8", qcache=qc)
        out3 = ops.reduce_out3(buf)
        sc = torch.tensor(0.005, device=dev)
        t_m1 = time_fn(lambda: ops.scaled_mm8(gt8, qc[0], sc))
        t_m2 = time_fn(lambda: ops.scaled_mm8(g8, qc[2], sc))
        print(f"fp8 GEMM gt8@zi_q     : {t_m1:7.3f} ms")
        print(f"fp8 GEMM g8@zt_q      : {t_m2:7.3f} ms")
        go = torch.tensor(1.0, device=dev)
        t_bfg8 = time_fn(lambda: ops.siglip_bwd_from_g(
            zi_d, zt_d, loss_mod.t_prime.detach(), loss_mod.bias.detach(),
            go, out3, g8, gt8, quant="fp8", qcache=qc))
        print(f"bwd_from_g fp8 total  : {t_bfg8:7.3f} ms")

    # 6. normalize fwd+bwd alone
    x = torch.randn(b, d, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    go = torch.randn(b, d, device=dev, dtype=torch.bfloat16)

    def norm_fb():
        y = F.normalize(x, dim=-1)
        torch.autograd.grad(y, x, go)
    t_n = time_fn(norm_fb)
    print(f"F.normalize fwd+bwd   : {t_n:7.3f} ms  (×2 towers in the step)")


if __name__ == "__main__":
    main()
