cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
cd /tmp
timeout 240 rocprofv3 --kernel-trace -d /root/repo/gpurun_out/ktrA -- python /root/repo/bench.py --steps 8 --warmup 3 > /dev/null 2>&1
timeout 240 rocprofv3 --kernel-trace -d /root/repo/gpurun_out/ktrB -- python /root/repo/bench.py --steps 8 --warmup 3 --quant fp8 > /dev/null 2>&1
cd /root/repo
dbA=$(find gpurun_out/ktrA -name '*.db' | head -1)
dbB=$(find gpurun_out/ktrB -name '*.db' | head -1)
timeout 120 python tools/kstats.py "$dbA" > gpurun_out/kstats_final_bf16.txt 2>&1
timeout 120 python tools/kstats.py "$dbB" > gpurun_out/kstats_final_fp8.txt 2>&1
rm -rf gpurun_out/ktrA gpurun_out/ktrB
timeout 240 python -m pytest tests -m gpu -q 2>&1 | tail -1
head -8 gpurun_out/kstats_final_bf16.txt
echo ---
head -8 gpurun_out/kstats_final_fp8.txt
