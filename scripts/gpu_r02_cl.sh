# channels_last A/B with NHWC-aware fused epilogues + cl kernel test.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 420 python -c "import __graft_entry__ as g; g.build()" > gpurun_out/r02cl_build.log 2>&1
echo "build rc=$?"
timeout 300 python -m pytest tests/test_hip_shm_gpu.py -q -k "channels_last or bias" > gpurun_out/r02cl_pytest.log 2>&1
echo "pytest rc=$?"; grep -E "passed|failed" gpurun_out/r02cl_pytest.log
timeout 300 python bench.py --steps 15 --warmup 5 > gpurun_out/r02cl_bench_nchw.log 2>&1
echo "nchw rc=$?"; tail -1 gpurun_out/r02cl_bench_nchw.log | head -c 200; echo
CLIENT_AMD_CHANNELS_LAST=1 timeout 300 python bench.py --steps 15 --warmup 5 > gpurun_out/r02cl_bench_cl.log 2>&1
echo "cl rc=$?"; tail -1 gpurun_out/r02cl_bench_cl.log | head -c 200; echo
echo DONE
