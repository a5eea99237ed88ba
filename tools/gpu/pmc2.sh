cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
cat > /tmp/pmc_driver2.py <<'PY'
import sys; sys.path.insert(0, "/root/repo")
import torch, torch.nn.functional as F
from distributed_sigmoid_loss_amd import ops
b, d = 32768, 768
zi = F.normalize(torch.randn(b, d, device="cuda"), dim=-1).bfloat16()
zt = F.normalize(torch.randn(b, d, device="cuda"), dim=-1).bfloat16()
tp = torch.tensor(2.302585, device="cuda"); bs = torch.tensor(-10.0, device="cuda")
qc = ops.quantize_fp8_pair(zi, zt)
go = torch.tensor(1.0, device="cuda")
for _ in range(3):
    ops.siglip_fwd(zi, zt, tp, bs, 0)
    buf, g, _ = ops.siglip_fwd_g(zi, zt, tp, bs, 0)
    ops.siglip_bwd_from_g(zi, zt, tp, bs, go, ops.reduce_out3(buf), g, None)
    b8, g8, gt8 = ops.siglip_fwd_g(zi, zt, tp, bs, 0, quant="fp8", qcache=qc)
    ops.siglip_bwd_from_g(zi, zt, tp, bs, go, ops.reduce_out3(b8), g8, gt8, quant="fp8", qcache=qc)
    y = ops.l2_normalize(zi.clone().requires_grad_(True))
torch.cuda.synchronize(); print("done")
PY
cd /tmp
timeout 400 rocprofv3 --pmc "MfmaUtil VALUBusy MemUnitStalled SQ_LDS_BANK_CONFLICT" --output-format csv -d /root/repo/gpurun_out/pmc2 -o pmc2 -- python /tmp/pmc_driver2.py > /root/repo/gpurun_out/pmc2.log 2>&1
echo "pmc rc=$?"
timeout 400 rocprofv3 --pmc "SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_WAVE_CYCLES" --output-format csv -d /root/repo/gpurun_out/pmc2 -o pmc2b -- python /tmp/pmc_driver2.py >> /root/repo/gpurun_out/pmc2.log 2>&1
echo "pmc2 rc=$?"
