cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
cat > /tmp/pmc_driver.py <<'PY'
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
torch.cuda.synchronize(); print("pmc driver done")
PY
cd /tmp
timeout 240 rocprofv3 --kernel-trace -d /root/repo/gpurun_out/prof_bf -- python /root/repo/bench.py --steps 8 --warmup 3 > /root/repo/gpurun_out/ptrace_bf16.log 2>&1
echo "trace_bf16 rc=$?"
timeout 240 rocprofv3 --kernel-trace -d /root/repo/gpurun_out/prof_f8 -- python /root/repo/bench.py --steps 8 --warmup 3 --quant fp8 > /root/repo/gpurun_out/ptrace_fp8.log 2>&1
echo "trace_fp8 rc=$?"
timeout 300 rocprofv3 --pmc "MfmaUtil VALUBusy MemUnitStalled SQ_LDS_BANK_CONFLICT" --output-format csv -d /root/repo/gpurun_out/pmc -o pmc_r2 -- python /tmp/pmc_driver.py > /root/repo/gpurun_out/pmc_run.log 2>&1
echo "pmc rc=$?"
cd /root/repo
for db in gpurun_out/prof_bf/*/*.db; do timeout 120 python tools/kstats.py "$db" > gpurun_out/kstats_r2_bf16.txt 2>&1; done
for db in gpurun_out/prof_f8/*/*.db; do timeout 120 python tools/kstats.py "$db" > gpurun_out/kstats_r2_fp8.txt 2>&1; done
rm -rf gpurun_out/prof_bf gpurun_out/prof_f8
timeout 300 python bench.py --steps 5 --warmup 2 --global-batch 131072 --dim 1024 > gpurun_out/bench_131k.log 2>&1
echo "131k rc=$?"
timeout 300 python examples/train_siglip.py --steps 20 --batch-per-gpu 8192 > gpurun_out/example_run.log 2>&1
echo "example rc=$?"
tail -2 gpurun_out/bench_131k.log; tail -3 gpurun_out/example_run.log
