#!/usr/bin/env python3
"""300-step stability soak at the headline config: loss finiteness +
allocator high-water marks every 50 steps."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from distributed_sigmoid_loss_amd import DistributedSigmoidLoss
from distributed_sigmoid_loss_amd.models import TwoTowerModel

quant = sys.argv[1] if len(sys.argv) > 1 else "bf16"
b, d = 32768, 768
torch.manual_seed(0)
model = TwoTowerModel(d, d).to(device="cuda", dtype=torch.bfloat16)
loss_mod = DistributedSigmoidLoss(b, quant=quant).cuda()
opt = torch.optim.SGD(list(model.parameters()) + list(loss_mod.parameters()),
                      lr=1e-4)
img = torch.randn(b, d, device="cuda", dtype=torch.bfloat16)
txt = torch.randn(b, d, device="cuda", dtype=torch.bfloat16)
t0 = time.perf_counter()
N = int(sys.argv[2]) if len(sys.argv) > 2 else 300
for step in range(N):
    opt.zero_grad(set_to_none=True)
    zi, zt = model(img, txt)
    loss = loss_mod(zi, zt)
    loss.backward()
    opt.step()
    if (step + 1) % 100 == 0 or step == 49:
        torch.cuda.synchronize()
        v = float(loss.detach())
        assert v == v and v > 0, f"bad loss {v} at step {step}"
        print(f"step {step+1}: loss={v:.3f} "
              f"alloc={torch.cuda.memory_allocated()/2**30:.2f}GiB "
              f"peak={torch.cuda.max_memory_allocated()/2**30:.2f}GiB",
              flush=True)
el = time.perf_counter() - t0
print(f"{N} steps OK ({quant}), {el:.1f}s wall, {b*N/el/1e6:.2f}M pairs/s")
