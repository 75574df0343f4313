# Decode deep-dive: trace2 split (stage/submit/wait) + rocprof kernel
# stats of the bmm decode path + BERT regression check.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python -c "import __graft_entry__ as g; g.build()" > gpurun_out/r02d_build.log 2>&1
echo "build rc=$?"

# trace2: where do the 7.3 ms go (stage / submit / GPU wait)
CLIENT_AMD_DECODE_TRACE=2 timeout 420 python scripts/profile_decode.py > gpurun_out/r02d_trace2.log 2>&1
echo "trace2 rc=$?"
grep "decode-trace2" gpurun_out/r02d_trace2.log | tail -4

# rocprof kernel stats of the same engine run
export TMPDIR=/tmp
mkdir -p gpurun_out/prof_decode_r02
(cd /tmp && timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_decode_r02 -- \
  python /root/repo/scripts/profile_decode.py > /root/repo/gpurun_out/r02d_prof.log 2>&1)
echo "prof rc=$?"
tail -2 gpurun_out/r02d_prof.log
for db in gpurun_out/prof_decode_r02/*/*_results.db; do
  python scripts/summarize_rocpd.py "$db" 2>/dev/null | head -18
done

# BERT-large seq128 regression check (round-2 changes shouldn't move it)
python -m client_amd.server --models bert_large --grpc-port 18003 --dynamic-batching > gpurun_out/r02d_bert_server.log 2>&1 &
B=$!
for i in $(seq 120); do grep -q GRPC_READY gpurun_out/r02d_bert_server.log && break; sleep 2; done
timeout 300 python -m client_amd.perf -m bert_large -u 127.0.0.1:18003 -i grpc -b 8 \
  --shared-memory cuda --concurrency-range 8:8:1 --measurement-interval 2 \
  --warmup 4 --max-windows 3 --json gpurun_out/r02d_bert.json > gpurun_out/r02d_bert.log 2>&1
echo "bert rc=$?"
python -c "
import json
for r in json.load(open('gpurun_out/r02d_bert.json')): print('bert c',r['concurrency'],r['inferences_per_sec'],'p99us',r['latency_us']['p99'])" || true
kill $B 2>/dev/null; wait $B 2>/dev/null
echo DONE
