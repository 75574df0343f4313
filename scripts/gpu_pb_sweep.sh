# Validate the new gRPC HIP-shm hipcc test + sweep server preferred
# batch size in the bench.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 600 python -m pytest tests/test_cpp_client.py -m gpu -q > gpurun_out/pb_pytest.log 2>&1
echo "pytest rc=$?"
tail -2 gpurun_out/pb_pytest.log

for pb in 48 64; do
  timeout 300 python bench.py --steps 10 --warmup 3 --preferred-batch-size $pb \
    > gpurun_out/pb_bench_$pb.log 2>&1
  echo "bench pb=$pb rc=$?"
  tail -1 gpurun_out/pb_bench_$pb.log | head -c 300
  echo
done
echo DONE
