set -x
export TMPDIR=/tmp
cd /tmp
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p /root/repo/gpurun_out/prof_decode
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_decode -- \
  python /root/repo/scripts/profile_decode.py > /root/repo/gpurun_out/prof_decode.log 2>&1
echo "prof rc=$?"
tail -3 /root/repo/gpurun_out/prof_decode.log
ls /root/repo/gpurun_out/prof_decode
# keep only the stats csv (trace files can be big)
find /root/repo/gpurun_out/prof_decode -name "*kernel_trace*" -size +5M -delete 2>/dev/null
echo DONE
