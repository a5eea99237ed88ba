cd /root/repo
mkdir -p gpurun_out
export PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1
for d in 1024 1152 1536; do
  PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tune_d${d}_.csv timeout 500 python bench.py --steps 8 --warmup 3 --quant fp8 --dim $d > /dev/null 2>&1
  echo "tuned d=$d rc=$?"
done
unset PYTORCH_TUNABLEOP_ENABLED
ls gpurun_out/tune_d*
