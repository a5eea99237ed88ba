set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -3 > gpurun_out/pytest7.log
python tools/probe_step.py fp8 > gpurun_out/pstep_fp8.log 2>&1
export TMPDIR=/tmp; cd /tmp
rocprofv3 --kernel-trace -d /root/repo/gpurun_out/prof_r2 -o r2bf16 -- python /root/repo/bench.py --steps 10 --warmup 4 > /root/repo/gpurun_out/prof_bf16.log 2>&1
rocprofv3 --kernel-trace -d /root/repo/gpurun_out/prof_r2 -o r2fp8 -- python /root/repo/bench.py --steps 10 --warmup 4 --quant fp8 > /root/repo/gpurun_out/prof_fp8.log 2>&1
cd /root/repo
for db in gpurun_out/prof_r2/*r2bf16*.db; do python tools/kstats.py "$db" > gpurun_out/kstats_bf16.txt; done
for db in gpurun_out/prof_r2/*r2fp8*.db; do python tools/kstats.py "$db" > gpurun_out/kstats_fp8.txt; done
rm -rf gpurun_out/prof_r2
echo done
