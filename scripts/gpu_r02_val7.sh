# Validation 7: confirm full recovery with device-sync default +
# GQA-bmm + pinned staging: ResNet ~8.5k, DenseNet ~4.5k+, llama
# c8 + long512 final numbers.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python -c "import __graft_entry__ as g; g.build()" > gpurun_out/r02v7_build.log 2>&1
echo "build rc=$?"

timeout 300 python bench.py --steps 20 --warmup 5 > gpurun_out/r02v7_bench.log 2>&1
echo "bench rc=$?"; tail -1 gpurun_out/r02v7_bench.log | head -c 260; echo

python -m client_amd.server --models densenet121 --grpc-port 18005 --dynamic-batching > gpurun_out/r02v7_dn_server.log 2>&1 &
S1=$!
for i in $(seq 90); do grep -q GRPC_READY gpurun_out/r02v7_dn_server.log && break; sleep 2; done
timeout 300 python -m client_amd.perf -m densenet121 -u 127.0.0.1:18005 -i grpc -b 8 \
  --shared-memory cuda --concurrency-range 4:8:4 --measurement-interval 2 \
  --warmup 4 --max-windows 4 --json gpurun_out/r02v7_densenet.json > gpurun_out/r02v7_dn.log 2>&1
echo "densenet rc=$?"
python -c "
import json
for r in json.load(open('gpurun_out/r02v7_densenet.json')): print('c',r['concurrency'],r['inferences_per_sec'],'p99us',r['latency_us']['p99'],'server',r.get('server'))" || true
kill $S1 2>/dev/null; wait $S1 2>/dev/null

python -m client_amd.server --models llama3_8b --grpc-port 18001 > gpurun_out/r02v7_ll_server.log 2>&1 &
SRV=$!
for i in $(seq 120); do grep -q GRPC_READY gpurun_out/r02v7_ll_server.log && break; sleep 2; done
timeout 420 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 8 --requests 16 --prompt-tokens 128 --output-tokens 256 \
  --json gpurun_out/r02v7_genai_c8.json > gpurun_out/r02v7_g1.log 2>&1
echo "genai c8 rc=$?"
timeout 600 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 8 --requests 16 --prompt-tokens 512 --output-tokens 512 \
  --json gpurun_out/r02v7_genai_long512_c8.json > gpurun_out/r02v7_g2.log 2>&1
echo "genai long rc=$?"
timeout 300 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 1 --requests 4 --prompt-tokens 128 --output-tokens 256 \
  --json gpurun_out/r02v7_genai_c1.json > gpurun_out/r02v7_g0.log 2>&1
echo "genai c1 rc=$?"
for f in gpurun_out/r02v7_genai_c8.json gpurun_out/r02v7_genai_long512_c8.json gpurun_out/r02v7_genai_c1.json; do
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

timeout 600 python -m pytest tests/test_hip_shm_gpu.py -q > gpurun_out/r02v7_pytest.log 2>&1
echo "pytest rc=$?"; grep -E "passed|failed" gpurun_out/r02v7_pytest.log
echo DONE
