cd /root/repo
mkdir -p gpurun_out
{
echo "=== kernels ==="; timeout 180 python -m pytest tests/test_gpu_kernels.py -q 2>&1 | tail -5
echo "=== savedg ==="; timeout 180 python -m pytest tests/test_gpu_savedg.py -q 2>&1 | tail -5
echo "=== fp8+fuzz ==="; timeout 180 python -m pytest tests/test_gpu_fp8.py tests/test_gpu_fuzz.py -q 2>&1 | tail -5
echo "=== multigpu (skips) ==="; timeout 180 python -m pytest tests/test_multigpu_rccl.py -q 2>&1 | tail -5
echo "=== big fp8 repro (serialized) ==="
timeout 240 python - <<'PY' 2>&1 | tail -12
import os
os.environ["AMD_SERIALIZE_KERNEL"] = "3"
import torch, torch.nn.functional as F
from distributed_sigmoid_loss_amd import ops
b, d = 32768, 768
zi = F.normalize(torch.randn(b, d, device="cuda"), dim=-1).bfloat16()
zt = F.normalize(torch.randn(b, d, device="cuda"), dim=-1).bfloat16()
tp = torch.tensor(2.302585, device="cuda"); bs = torch.tensor(-10.0, device="cuda")
qc = ops.quantize_fp8_pair(zi, zt)
buf, g8, gt8 = ops.siglip_fwd_g(zi, zt, tp, bs, 0, quant="fp8", qcache=qc)
torch.cuda.synchronize(); print("fwd_g fp8 ok", float(ops.reduce_out3(buf)[0]))
sc = torch.tensor(0.005, device="cuda")
r1 = ops.scaled_mm8(gt8, qc[0], sc); torch.cuda.synchronize(); print("mm8 gt ok", float(r1.abs().sum()))
r2 = ops.scaled_mm8(g8, qc[2], sc); torch.cuda.synchronize(); print("mm8 g ok", float(r2.abs().sum()))
out3 = ops.reduce_out3(buf)
go = torch.tensor(1.0, device="cuda")
o = ops.siglip_bwd_from_g(zi, zt, tp, bs, go, out3, g8, gt8, quant="fp8", qcache=qc)
torch.cuda.synchronize(); print("bwd_from_g fp8 ok")
PY
echo "=== big l2norm repro ==="
timeout 120 python - <<'PY' 2>&1 | tail -4
import os
os.environ["AMD_SERIALIZE_KERNEL"] = "3"
import torch
from distributed_sigmoid_loss_amd import ops
x = torch.randn(32768, 768, device="cuda", dtype=torch.bfloat16, requires_grad=True)
y = ops.l2_normalize(x); y.backward(torch.randn_like(y))
torch.cuda.synchronize(); print("l2norm ok", float(x.grad.abs().sum()))
PY
} > gpurun_out/diag1.log 2>&1
cat gpurun_out/diag1.log
