# Round-2 kernel validation: fused bottleneck numerics + A/B, tiled
# transpose bandwidth, batched preprocess.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python -c "import __graft_entry__ as g; g.build()" > gpurun_out/r02k_build.log 2>&1
echo "build rc=$?"

timeout 600 python -m pytest tests/test_hip_shm_gpu.py -q > gpurun_out/r02k_pytest.log 2>&1
echo "pytest rc=$?"
tail -3 gpurun_out/r02k_pytest.log

timeout 420 python scripts/profile_kernels.py > gpurun_out/r02k_kernels.log 2>&1
echo "kernels rc=$?"
cat gpurun_out/r02k_kernels.log

# A/B: fused epilogues (default) vs plain fold
timeout 300 python bench.py --steps 20 --warmup 5 > gpurun_out/r02k_bench_fused.log 2>&1
echo "bench fused rc=$?"
tail -1 gpurun_out/r02k_bench_fused.log
CLIENT_AMD_FUSED_BIAS=0 timeout 300 python bench.py --steps 20 --warmup 5 > gpurun_out/r02k_bench_plain.log 2>&1
echo "bench plain rc=$?"
tail -1 gpurun_out/r02k_bench_plain.log
echo DONE
