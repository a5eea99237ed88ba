#!/usr/bin/env python3
"""Flag-matrix sweep for the fused kernels (round 2, post atomic fix):
GROUP_M × XCD remap × non-temporal g stores, on fwd / fwd+g / bwd-g.

Run on a GPU box:  python tools/sweep_flags.py [--batch 32768] [--dim 768]
"""

from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from distributed_sigmoid_loss_amd import ops


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
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=32768)
    p.add_argument("--dim", type=int, default=768)
    args = p.parse_args()
    b, d = args.batch, args.dim

    torch.manual_seed(0)
    zi = F.normalize(torch.randn(b, d, device="cuda"), dim=-1).bfloat16()
    zt = F.normalize(torch.randn(b, d, device="cuda"), dim=-1).bfloat16()
    tp = torch.tensor(2.302585, device="cuda")
    bs = torch.tensor(-10.0, device="cuda")
    tf = 2.0 * b * b * d / 1e9

    def setenv(xcd, gm, nt):
        os.environ["SIGLIP_XCD_SWZ"] = str(xcd)
        os.environ["SIGLIP_GROUP_M"] = str(gm)
        os.environ["SIGLIP_NT_G"] = str(nt)

    print(f"B={b} d={d}  (TF numbers on the bf16-flop basis)")
    print("-- fwd (MODE0) --")
    for xcd in (1, 0):
        for gm in (4, 8, 1, 16):
            setenv(xcd, gm, 0)
            t = time_fn(lambda: ops.siglip_fwd(zi, zt, tp, bs, 0))
            print(f"xcd={xcd} gm={gm:2d}        : {t:7.3f} ms  {tf/t:7.1f} TF")
    print("-- fwd+g bf16 (MODE2) --")
    for gm in (4, 8):
        for nt in (0, 1):
            setenv(1, gm, nt)
            t = time_fn(lambda: ops.siglip_fwd_g(zi, zt, tp, bs, 0))
            print(f"gm={gm:2d} nt={nt}         : {t:7.3f} ms  {tf/t:7.1f} TF")
    print("-- fwd+g fp8 --")
    qc = ops.quantize_fp8_pair(zi, zt)
    for nt in (0, 1):
        setenv(1, 4, nt)
        t = time_fn(lambda: ops.siglip_fwd_g(zi, zt, tp, bs, 0, quant="fp8",
                                             qcache=qc))
        print(f"nt={nt} (quant cached): {t:7.3f} ms  {tf/t:7.1f} TF")
    print("-- fwd+g mixed --")
    for nt in (0, 1):
        setenv(1, 4, nt)
        t = time_fn(lambda: ops.siglip_fwd_g(zi, zt, tp, bs, 0,
                                             quant="mixed", qcache=qc))
        print(f"nt={nt}              : {t:7.3f} ms  {tf/t:7.1f} TF")
    print("-- fp8 row-wise decomposition --")
    setenv(1, 4, 0)
    t_qrw = time_fn(lambda: ops.quantize_fp8_rowwise_pair(zi, zt))
    print(f"rowwise quant pair   : {t_qrw:7.3f} ms")
    qrw = ops.quantize_fp8_rowwise_pair(zi, zt)
    t_frw = time_fn(lambda: ops.siglip_fwd_g(zi, zt, tp, bs, 0, quant="fp8",
                                             qcache=qrw))
    print(f"fwd+g fp8 rowwise    : {t_frw:7.3f} ms  {tf/t_frw:7.1f} TF")
    buf, g8, gt8 = ops.siglip_fwd_g(zi, zt, tp, bs, 0, quant="fp8",
                                    qcache=qrw)
    out3 = ops.reduce_out3(buf)
    go = torch.tensor(1.0, device="cuda")
    t_brw = time_fn(lambda: ops.siglip_bwd_from_g(
        zi, zt, tp, bs, go, out3, g8, gt8, quant="fp8", qcache=qrw))
    print(f"bwd_from_g rowwise   : {t_brw:7.3f} ms")
    print("-- bwd recompute g-kernel full slab (MODE1) --")
    go = torch.tensor(1.0, device="cuda")
    for nt in (0, 1):
        setenv(1, 4, nt)
        t = time_fn(lambda: ops.siglip_bwd(zi, zt, tp, bs, 0, go, None))
        print(f"nt={nt} bwd total    : {t:7.3f} ms")
    setenv(1, 4, 0)
    # correctness spot-check of nt stores
    buf0, g0, _ = ops.siglip_fwd_g(zi, zt, tp, bs, 0)
    os.environ["SIGLIP_NT_G"] = "1"
    buf1, g1, _ = ops.siglip_fwd_g(zi, zt, tp, bs, 0)
    os.environ["SIGLIP_NT_G"] = "0"
    torch.cuda.synchronize()
    assert torch.equal(g0, g1), "nt stores changed g values!"
    print("nt-store correctness: OK")


if __name__ == "__main__":
    main()
