set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python bench.py --steps 30 --warmup 5 > gpurun_out/last_bench30.log 2>&1
echo "bench30 rc=$?"; tail -1 gpurun_out/last_bench30.log | head -c 250; echo

python -m client_amd.server --models densenet121 --grpc-port 18005 --dynamic-batching > gpurun_out/last_dn_server.log 2>&1 &
S1=$!
for i in $(seq 90); do grep -q GRPC_READY gpurun_out/last_dn_server.log && break; sleep 2; done
timeout 300 python -m client_amd.perf -m densenet121 -u 127.0.0.1:18005 -i grpc -b 8 \
  --shared-memory cuda --concurrency-range 4:8:4 --measurement-interval 2 \
  --warmup 4 --max-windows 4 --json gpurun_out/last_densenet.json > gpurun_out/last_dn.log 2>&1
echo "densenet rc=$?"
python -c "
import json
for r in json.load(open('gpurun_out/last_densenet.json')): print('c',r['concurrency'],r['inferences_per_sec'],'p99us',r['latency_us']['p99'],'server',r.get('server'))" || true
kill $S1 2>/dev/null; wait $S1 2>/dev/null

python -m client_amd.server --models llama3_8b --grpc-port 18001 > gpurun_out/last_ll_server.log 2>&1 &
S2=$!
for i in $(seq 90); do grep -q GRPC_READY gpurun_out/last_ll_server.log && break; sleep 2; done
timeout 420 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 8 --requests 6 --prompt-tokens 256 --output-tokens 512 \
  --json gpurun_out/last_llama_long.json > gpurun_out/last_ll.log 2>&1
echo "llama rc=$?"
python -c "
import json; d=json.load(open('gpurun_out/last_llama_long.json'))
print('tok/s',d['output_tokens_per_sec'],'ITL',d['inter_token_latency_ms'],'TTFT',d['ttft_ms'],'err',d['errors'])" || true
kill $S2 2>/dev/null; wait $S2 2>/dev/null
echo DONE
