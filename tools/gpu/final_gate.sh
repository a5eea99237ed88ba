cd /root/repo
mkdir -p gpurun_out
{
timeout 120 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -1
for i in 1 2; do timeout 240 python -m pytest tests -m gpu -q 2>&1 | tail -1; done
timeout 400 python tools/fuzz_campaign.py 240 20260917 2>&1 | tail -1
echo "=== bf16 ==="; timeout 120 python bench.py --steps 40 --warmup 10 2>/dev/null
echo "=== fp8 ==="; timeout 120 python bench.py --steps 40 --warmup 10 --quant fp8 2>/dev/null
echo "=== mixed ==="; timeout 120 python bench.py --steps 40 --warmup 10 --quant mixed 2>/dev/null
echo "=== floor ==="; timeout 200 python bench.py --steps 10 --warmup 3 --impl torch 2>/dev/null
echo "=== 131k ==="; timeout 300 python bench.py --steps 5 --warmup 2 --global-batch 131072 --dim 1024 2>/dev/null
echo "=== fp8 d1152 ==="; timeout 200 python bench.py --steps 20 --warmup 5 --quant fp8 --dim 1152 2>/dev/null
} > gpurun_out/final_gate.log 2>&1
grep -E 'smoke|passed|/|===' gpurun_out/final_gate.log | grep -oE 'smoke ok.*|[0-9]+ passed.*|[0-9]+/[0-9]+ passed|=== [a-z0-9 ]+===' 
grep -oE '"value": [0-9.]+' gpurun_out/final_gate.log
