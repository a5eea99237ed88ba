set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -6 > gpurun_out/pytest3.log
{
echo "=== bf16 savedg ==="
python bench.py --steps 30 --warmup 8
echo "=== bf16 recompute ==="
SIGLIP_SAVE_G=0 python bench.py --steps 30 --warmup 8
echo "=== mixed ==="
python bench.py --steps 30 --warmup 8 --quant mixed
echo "=== fp8 ==="
python bench.py --steps 30 --warmup 8 --quant fp8
echo "=== bf16 + graph ==="
python bench.py --steps 30 --warmup 8 --graph
echo "=== bf16 all_gather ==="
python bench.py --steps 30 --warmup 8 --strategy all_gather
echo "=== bf16 torch floor ==="
python bench.py --steps 10 --warmup 3 --impl torch
} > gpurun_out/bench3.log 2>&1
echo done
