#!/usr/bin/env python3
"""Per-phase timing decomposition of the flagship step on one GPU.

Times each component with CUDA events: fwd kernel, bwd g-kernel, the two
gradient GEMMs, encoder fwd/bwd, optimizer.  Run on a GPU box:
    python tools/perf_probe.py [--batch 32768] [--dim 768]
"""

from __future__ import annotations

import argparse
import ctypes
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
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=32768)
    p.add_argument("--dim", type=int, default=768)
    p.add_argument("--col-chunk", type=int, default=8192)
    args = p.parse_args()
    b, d = args.batch, args.dim

    torch.manual_seed(0)
    dev = "cuda"
    zi = F.normalize(torch.randn(b, d, device=dev), dim=-1).bfloat16()
    zt = F.normalize(torch.randn(b, d, device=dev), dim=-1).bfloat16()
    tp = torch.tensor(2.302585, device=dev)
    bs = torch.tensor(-10.0, device=dev)
    go = torch.tensor(1.0, device=dev)

    flops_fwd = 2.0 * b * b * d

    t_fwd = time_fn(lambda: ops.siglip_fwd(zi, zt, tp, bs, 0))
    print(f"fwd fused kernel      : {t_fwd:8.3f} ms   "
          f"{flops_fwd / t_fwd / 1e9:7.1f} TF/s")

    t_fwd8 = time_fn(lambda: ops.siglip_fwd(zi, zt, tp, bs, 0, quant="fp8"))
    print(f"fwd fused fp8 (MX)    : {t_fwd8:8.3f} ms   "
          f"{flops_fwd / t_fwd8 / 1e9:7.1f} TF/s  (incl quantize)")

    go1 = torch.tensor(1.0, device=dev)
    t_bwd8 = time_fn(lambda: ops.siglip_bwd(zi, zt, tp, bs, 0, go1, None,
                                            quant="fp8"))
    print(f"bwd total fp8 (MX)    : {t_bwd8:8.3f} ms")

    # rocBLAS ceiling reference: the same-shape plain GEMM (materializes the
    # logits we fuse away) — an upper bound on achievable MFMA throughput.
    ref_n = min(b, 8192)
    zt_ref = zt[:ref_n].contiguous()
    t_ref = time_fn(lambda: zi @ zt_ref.T)
    print(f"rocBLAS zi@zt.T ref   : {t_ref:8.3f} ms   "
          f"{2.0 * b * ref_n * d / t_ref / 1e9:7.1f} TF/s  (n={ref_n})")

    # bwd pieces: g-kernel alone, then GEMMs alone.
    lib = ops._require_lib()
    c = min(args.col_chunk, b)
    g = torch.empty((b, c), device=dev, dtype=torch.bfloat16)
    scal = ops._out_buf(dev)
    ztc = zt[:c].contiguous()
    stream = torch.cuda.current_stream().cuda_stream

    def g_kernel():
        ops._check(lib.siglip_bwd_g_bf16(
            ctypes.c_void_p(stream),
            ctypes.c_void_p(zi.data_ptr()), ctypes.c_void_p(ztc.data_ptr()),
            ctypes.c_void_p(tp.data_ptr()), ctypes.c_void_p(bs.data_ptr()),
            ctypes.c_void_p(g.data_ptr()), ctypes.c_void_p(scal.data_ptr()),
            b, c, d, 0, ops._kernel_flags()), "bwd_g")

    t_g = time_fn(g_kernel)
    nchunks = (b + c - 1) // c
    print(f"bwd g-kernel ({nchunks}x{c:6d}): {t_g * nchunks:8.3f} ms   "
          f"{2.0 * b * c * d / t_g / 1e9:7.1f} TF/s per-chunk")

    t_mm1 = time_fn(lambda: g @ ztc)
    t_mm2 = time_fn(lambda: g.T @ zi)
    print(f"GEMM g@zt   ({nchunks} chunks): {t_mm1 * nchunks:8.3f} ms   "
          f"{2.0 * b * c * d / t_mm1 / 1e9:7.1f} TF/s")
    print(f"GEMM gT@zi  ({nchunks} chunks): {t_mm2 * nchunks:8.3f} ms   "
          f"{2.0 * b * c * d / t_mm2 / 1e9:7.1f} TF/s")

    t_bwd = time_fn(lambda: ops.siglip_bwd(zi, zt, tp, bs, 0, go,
                                           args.col_chunk))
    print(f"bwd total (wrapper)   : {t_bwd:8.3f} ms")

    # fwd+g (saved-g) pieces: the fused fwd emits loss + g slab + scalars;
    # backward is two full-slab GEMMs.
    t_fwdg = time_fn(lambda: ops.siglip_fwd_g(zi, zt, tp, bs, 0))
    print(f"fwd+g fused kernel    : {t_fwdg:8.3f} ms   "
          f"{flops_fwd / t_fwdg / 1e9:7.1f} TF/s")
    buf3, gfull, _ = ops.siglip_fwd_g(zi, zt, tp, bs, 0)
    out3 = ops.reduce_out3(buf3)
    t_bfg = time_fn(lambda: ops.siglip_bwd_from_g(zi, zt, tp, bs, go, out3,
                                                  gfull, None))
    print(f"bwd_from_g (2 GEMMs)  : {t_bfg:8.3f} ms")
    t_mmA = time_fn(lambda: gfull @ zt)
    t_mmB = time_fn(lambda: gfull.T @ zi)
    print(f"GEMM g@zt  full slab  : {t_mmA:8.3f} ms   "
          f"{2.0 * b * b * d / t_mmA / 1e9:7.1f} TF/s")
    print(f"GEMM gT@zi full slab  : {t_mmB:8.3f} ms   "
          f"{2.0 * b * b * d / t_mmB / 1e9:7.1f} TF/s")
    del buf3, out3, gfull
    t_fwdg8 = time_fn(lambda: ops.siglip_fwd_g(zi, zt, tp, bs, 0,
                                               quant="fp8"))
    print(f"fwd+g fp8 (incl quant): {t_fwdg8:8.3f} ms   "
          f"{flops_fwd / t_fwdg8 / 1e9:7.1f} TF/s")
    t_fwdgm = time_fn(lambda: ops.siglip_fwd_g(zi, zt, tp, bs, 0,
                                               quant="mixed"))
    print(f"fwd+g mixed           : {t_fwdgm:8.3f} ms   "
          f"{flops_fwd / t_fwdgm / 1e9:7.1f} TF/s")

    # Whole training step as the bench runs it.
    model = TwoTowerModel(d, d).to(device=dev, dtype=torch.bfloat16)
    loss_mod = DistributedSigmoidLoss(b, col_chunk=args.col_chunk).cuda()
    opt = torch.optim.SGD(list(model.parameters()) +
                          list(loss_mod.parameters()), lr=1e-4)
    img = torch.randn(b, d, device=dev, dtype=torch.bfloat16)
    txt = torch.randn(b, d, device=dev, dtype=torch.bfloat16)

    def full_step():
        opt.zero_grad(set_to_none=True)
        a, t_ = model(img, txt)
        loss = loss_mod(a, t_)
        loss.backward()
        opt.step()

    t_step = time_fn(full_step, iters=5, warmup=2)
    print(f"full train step       : {t_step:8.3f} ms   "
          f"({b / t_step * 1000:,.0f} pairs/s)")


if __name__ == "__main__":
    main()
