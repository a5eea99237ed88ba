cd /root/repo
mkdir -p gpurun_out
{
for i in 1 2; do timeout 240 python -m pytest tests -m gpu -q 2>&1 | tail -1; done
timeout 400 python tools/fuzz_campaign.py 240 31415 2>&1 | tail -1
timeout 300 python tools/soak.py bf16 2000 2>&1 | tail -1
timeout 150 python tools/probe_step.py bf16 2>&1 | grep -E 'encode|loss|full'
timeout 150 python tools/probe_step.py fp8 2>&1 | grep -E 'loss|quantize'
} > gpurun_out/gate1.log 2>&1
{
echo "=== bf16 ==="; timeout 120 python bench.py --steps 40 --warmup 10
echo "=== mixed ==="; timeout 120 python bench.py --steps 40 --warmup 10 --quant mixed
echo "=== fp8 ==="; timeout 120 python bench.py --steps 40 --warmup 10 --quant fp8
echo "=== floor ==="; timeout 180 python bench.py --steps 10 --warmup 3 --impl torch
echo "=== 131k ==="; timeout 300 python bench.py --steps 5 --warmup 2 --global-batch 131072 --dim 1024
} > gpurun_out/gate1_bench.log 2>&1
cat gpurun_out/gate1.log
grep -oE '=== [a-z0-9 ]+===|\"value\": [0-9.]+' gpurun_out/gate1_bench.log
