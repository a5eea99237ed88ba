set -x
cd /root/repo
mkdir -p gpurun_out
{
echo "=== bf16 ==="; python bench.py --steps 30 --warmup 8
echo "=== fp8 ==="; python bench.py --steps 30 --warmup 8 --quant fp8
echo "=== mixed ==="; python bench.py --steps 30 --warmup 8 --quant mixed
echo "=== bf16 csv ==="; python bench.py --steps 20 --warmup 5 --csv gpurun_out/phases_bf16.csv
echo "=== fp8 csv ==="; python bench.py --steps 20 --warmup 5 --quant fp8 --csv gpurun_out/phases_fp8.csv
} > gpurun_out/bench5.log 2>&1
tail -2 gpurun_out/phases_bf16.csv
echo done
