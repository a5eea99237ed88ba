cd /root/repo
mkdir -p gpurun_out
{
for i in 1 2 3; do timeout 240 python -m pytest tests -m gpu -q 2>&1 | tail -1; done
} > gpurun_out/race_screen_r2.log 2>&1
timeout 500 python tools/fuzz_campaign.py 300 20260914 > gpurun_out/fuzz_final.log 2>&1
timeout 300 python tools/soak.py bf16 > gpurun_out/soak_r2.log 2>&1
timeout 300 python tools/soak.py fp8 >> gpurun_out/soak_r2.log 2>&1
{
echo "=== bf16 ==="; timeout 120 python bench.py --steps 30 --warmup 8
echo "=== mixed ==="; timeout 120 python bench.py --steps 30 --warmup 8 --quant mixed
echo "=== fp8 ==="; timeout 120 python bench.py --steps 30 --warmup 8 --quant fp8
echo "=== fp8 rowwise ==="; SIGLIP_FP8_ROWWISE=1 timeout 120 python bench.py --steps 30 --warmup 8 --quant fp8
echo "=== bf16 all_gather ==="; timeout 120 python bench.py --steps 30 --warmup 8 --strategy all_gather
echo "=== bf16 recompute ==="; SIGLIP_SAVE_G=0 timeout 120 python bench.py --steps 30 --warmup 8
echo "=== torch floor ==="; timeout 180 python bench.py --steps 10 --warmup 3 --impl torch
echo "=== config1 b=8192 ==="; timeout 120 python bench.py --steps 30 --warmup 8 --global-batch 8192
echo "=== config5-shape fp8 d=1152 ==="; timeout 120 python bench.py --steps 20 --warmup 5 --quant fp8 --dim 1152
echo "=== config4-shape 131k d=1024 ==="; timeout 300 python bench.py --steps 5 --warmup 2 --global-batch 131072 --dim 1024
} > gpurun_out/bench_final_r2.log 2>&1
timeout 240 python tools/perf_probe.py > gpurun_out/perf_probe_final_r2.log 2>&1
cat gpurun_out/race_screen_r2.log
tail -1 gpurun_out/fuzz_final.log
tail -2 gpurun_out/soak_r2.log
grep -oE '=== [a-z0-9 =_.-]+===|"value": [0-9.]+' gpurun_out/bench_final_r2.log
