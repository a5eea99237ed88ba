set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -8 > gpurun_out/pytest1.log
{
echo "=== bench bf16 savedg (default) ==="
python bench.py --steps 30 --warmup 8
echo "=== bench bf16 recompute (SIGLIP_SAVE_G=0) ==="
SIGLIP_SAVE_G=0 python bench.py --steps 30 --warmup 8
echo "=== bench mixed savedg ==="
python bench.py --steps 30 --warmup 8 --quant mixed
echo "=== bench fp8 savedg ==="
python bench.py --steps 30 --warmup 8 --quant fp8
echo "=== bench bf16 savedg + graph ==="
python bench.py --steps 30 --warmup 8 --graph
echo "=== all_gather strategy savedg ==="
python bench.py --steps 30 --warmup 8 --strategy all_gather
} > gpurun_out/bench1.log 2>&1
echo done
