set -x
export TMPDIR=/tmp
cd /tmp
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p /root/repo/gpurun_out/prof_serving
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_serving -- \
  python /root/repo/bench.py --steps 5 --warmup 2 > /root/repo/gpurun_out/prof_serving.log 2>&1
echo "rc=$?"
tail -2 /root/repo/gpurun_out/prof_serving.log
ls -la /root/repo/gpurun_out/prof_serving/*/ | head -6
echo DONE
