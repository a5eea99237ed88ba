cd /root/repo
mkdir -p gpurun_out
timeout 120 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/smoke_final.log 2>&1
echo "smoke rc=$?"
timeout 240 python -m pytest tests -m gpu -q 2>&1 | tail -1
{ echo bf16; timeout 120 python bench.py --steps 30 --warmup 8; echo fp8; timeout 120 python bench.py --steps 30 --warmup 8 --quant fp8; } > gpurun_out/bench_final2.log 2>&1
export TMPDIR=/tmp; cd /tmp
timeout 240 rocprofv3 --kernel-trace -d /root/repo/gpurun_out/ktr -o fin -- python /root/repo/bench.py --steps 8 --warmup 3 --quant fp8 > /dev/null 2>&1
cd /root/repo
for db in gpurun_out/ktr/*/*.db; do timeout 120 python tools/kstats.py "$db" > gpurun_out/kstats_final_fp8.txt 2>&1; done
rm -rf gpurun_out/ktr
tail -2 gpurun_out/smoke_final.log
grep -oE 'bf16|fp8|\"ms_per_step\": [0-9.]+' gpurun_out/bench_final2.log
head -6 gpurun_out/kstats_final_fp8.txt
