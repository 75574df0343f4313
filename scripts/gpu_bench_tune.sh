# Validate hipcc-link fix in GPU pytest + tune bench step structure.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 900 python -m pytest tests -m gpu -q > gpurun_out/tune_pytest.log 2>&1
echo "pytest rc=$?"
tail -2 gpurun_out/tune_pytest.log

for cfg in "16 8" "48 8" "48 12" "96 8"; do
  set -- $cfg
  timeout 300 python bench.py --steps 10 --warmup 3 --reqs-per-step $1 --concurrency $2 \
    > gpurun_out/tune_bench_r$1_c$2.log 2>&1
  echo "bench r$1 c$2 rc=$?"
  tail -1 gpurun_out/tune_bench_r$1_c$2.log | head -c 400
  echo
done
echo DONE
