set -x
cd /root/repo
mkdir -p gpurun_out
python tools/perf_probe.py > gpurun_out/probe2.log 2>&1
export TMPDIR=/tmp
cd /tmp
rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_savedg -- python /root/repo/bench.py --steps 10 --warmup 4 > /root/repo/gpurun_out/b_stats.log 2>&1
echo done
