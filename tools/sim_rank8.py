#!/usr/bin/env python3
"""Simulate the per-rank COMPUTE of the N=8 ring step on one GPU (no comm):
8 chunked fwd+g calls into the (b, W*b) slab + the saved-g backward +
towers + optimizer.  The gap between this and SCALE_r02's N=8 number is
the un-hidden communication + multi-process overhead."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, torch.nn.functional as F
from distributed_sigmoid_loss_amd import ops, DistributedSigmoidLoss
from distributed_sigmoid_loss_amd.models import TwoTowerModel

W, b, d = 8, 4096, 768
n = W * b
dev = "cuda"
torch.manual_seed(0)
model = TwoTowerModel(d, d).to(device=dev, dtype=torch.bfloat16)
opt = torch.optim.SGD(model.parameters(), lr=1e-4)
img = torch.randn(b, d, device=dev, dtype=torch.bfloat16)
txt = torch.randn(b, d, device=dev, dtype=torch.bfloat16)
tp = torch.tensor(2.302585, device=dev)
bs = torch.tensor(-10.0, device=dev)
chunks = [F.normalize(torch.randn(b, d, device=dev), dim=-1).bfloat16()
          for _ in range(W)]
go = torch.tensor(1.0, device=dev)

def step():
    opt.zero_grad(set_to_none=True)
    zi, zt = model(img, txt)
    zi = zi.detach().contiguous()
    g_slab = torch.empty((b, n), device=dev, dtype=torch.bfloat16)
    buf = ops._out_buf(dev)
    for src in range(W):
        ops.siglip_fwd_g(zi, chunks[src], tp, bs,
                         0 if src == 0 else None,
                         g_slab=g_slab, col0=src * b, out3=buf)
    out3 = ops.reduce_out3(buf)
    all_txt = torch.cat(chunks, dim=0)
    dzi, dzt, dtp, dbs = ops.siglip_bwd_from_g(zi, all_txt, tp, bs, go,
                                               out3, g_slab, None)
    zt.backward(dzt[:b] + torch.zeros_like(zt))
    opt.step()

def time_it(fn, iters=30, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000

ms = time_it(step)

# single-call variant (the all_gather strategy's compute shape)
all_txt0 = torch.cat(chunks, dim=0)

def step_single():
    opt.zero_grad(set_to_none=True)
    zi, zt = model(img, txt)
    zi = zi.detach().contiguous()
    buf, g_slab, _ = ops.siglip_fwd_g(zi, all_txt0, tp, bs, 0)
    out3 = ops.reduce_out3(buf)
    dzi, dzt, dtp, dbs = ops.siglip_bwd_from_g(zi, all_txt0, tp, bs, go,
                                               out3, g_slab, None)
    zt.backward(dzt[:b] + torch.zeros_like(zt))
    opt.step()

ms1 = time_it(step_single)
ideal = W * b / ms * 1000 / 1e6
print(f"simulated N=8 rank compute (8 ring chunks): {ms:.3f} ms/step")
print(f"simulated N=8 rank compute (single call)  : {ms1:.3f} ms/step")
print(f"-> comm-hidden whole-job ceiling {ideal:.1f}M pairs/s at 8 GPUs; "
      f"vs 8x linear of a 6.0ms N=1 step = {8 * 32768 / 6.0 / 1000:.1f}M "
      f"-> strong-scaling compute ceiling ~{6.0 / (8 * ms) * 100:.0f}%")
