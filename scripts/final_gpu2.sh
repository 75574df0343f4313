# Driver-mirror validation: build+smoke, full GPU pytest (incl. new
# gRPC HIP-shm hipcc test), default-config bench.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python -c "import __graft_entry__ as g; g.build(); g.smoke(); print('SMOKE-OK')" > gpurun_out/final2_entry.log 2>&1
echo "entry rc=$?"
tail -2 gpurun_out/final2_entry.log

timeout 1000 python -m pytest tests -m gpu -q > gpurun_out/final2_pytest.log 2>&1
echo "pytest rc=$?"
grep -E "passed|failed" gpurun_out/final2_pytest.log | tail -2

timeout 300 python bench.py --steps 10 --warmup 3 > gpurun_out/final2_bench.log 2>&1
echo "bench rc=$?"
tail -1 gpurun_out/final2_bench.log
echo DONE
