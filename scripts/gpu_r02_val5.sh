# Validation 5: default-stream revert (ResNet ~8.5k, DenseNet ~4.6k
# back) + decode-loop phase trace + fused-serving rocprof evidence.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python -c "import __graft_entry__ as g; g.build()" > gpurun_out/r02v5_build.log 2>&1
echo "build rc=$?"

timeout 300 python bench.py --steps 15 --warmup 5 > gpurun_out/r02v5_bench.log 2>&1
echo "bench rc=$?"; tail -1 gpurun_out/r02v5_bench.log | head -c 300; echo

python -m client_amd.server --models densenet121 --grpc-port 18005 --dynamic-batching > gpurun_out/r02v5_dn_server.log 2>&1 &
S1=$!
for i in $(seq 90); do grep -q GRPC_READY gpurun_out/r02v5_dn_server.log && break; sleep 2; done
timeout 300 python -m client_amd.perf -m densenet121 -u 127.0.0.1:18005 -i grpc -b 8 \
  --shared-memory cuda --concurrency-range 4:8:4 --measurement-interval 2 \
  --warmup 4 --max-windows 4 --json gpurun_out/r02v5_densenet.json > gpurun_out/r02v5_dn.log 2>&1
echo "densenet rc=$?"
python -c "
import json
for r in json.load(open('gpurun_out/r02v5_densenet.json')): print('c',r['concurrency'],r['inferences_per_sec'],'p99us',r['latency_us']['p99'],'server',r.get('server'))" || true
kill $S1 2>/dev/null; wait $S1 2>/dev/null

CLIENT_AMD_DECODE_TRACE=1 python -m client_amd.server --models llama3_8b --grpc-port 18001 > gpurun_out/r02v5_llama_server.log 2>&1 &
SRV=$!
for i in $(seq 120); do grep -q GRPC_READY gpurun_out/r02v5_llama_server.log && break; sleep 2; done
timeout 420 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 1 --requests 4 --prompt-tokens 128 --output-tokens 256 \
  --json gpurun_out/r02v5_genai_c1.json > gpurun_out/r02v5_genai0.log 2>&1
echo "genai c1 rc=$?"
timeout 420 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 8 --requests 16 --prompt-tokens 128 --output-tokens 256 \
  --json gpurun_out/r02v5_genai_c8.json > gpurun_out/r02v5_genai2.log 2>&1
echo "genai c8 rc=$?"
grep "decode-trace" gpurun_out/r02v5_llama_server.log | tail -8
for f in gpurun_out/r02v5_genai_c1.json gpurun_out/r02v5_genai_c8.json; do
  python - "$f" <<'PYEOF'
import json, sys
d = json.load(open(sys.argv[1]))
print(sys.argv[1].split('/')[-1], '-> tok/s', d['output_tokens_per_sec'],
      'ITL p50/p99/max', d['inter_token_latency_ms']['p50'],
      d['inter_token_latency_ms']['p99'], d['inter_token_latency_ms']['max'],
      'TTFT p50/p99', d['ttft_ms']['p50'], d['ttft_ms']['p99'],
      'err', d['errors'])
PYEOF
done
kill $SRV 2>/dev/null
wait $SRV 2>/dev/null

# rocprof of the fused serving path (elementwise cluster must be gone)
export TMPDIR=/tmp
mkdir -p gpurun_out/prof_fused
(cd /tmp && timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_fused -- \
  python /root/repo/bench.py --steps 5 --warmup 2 > /root/repo/gpurun_out/r02v5_prof.log 2>&1)
echo "rocprof rc=$?"
ls gpurun_out/prof_fused/ | head -4
echo DONE
