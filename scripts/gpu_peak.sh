set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 300 python bench.py --steps 10 --warmup 3 --concurrency 16 --reqs-per-step 96 > gpurun_out/peak_bench.log 2>&1
echo "peak rc=$?"; tail -1 gpurun_out/peak_bench.log | head -c 280; echo

python -m client_amd.server --models llama3_8b --grpc-port 18001 > gpurun_out/peak_ll_server.log 2>&1 &
SRV=$!
for i in $(seq 90); do grep -q GRPC_READY gpurun_out/peak_ll_server.log && break; sleep 2; done
timeout 300 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 4 --requests 8 --prompt-tokens 512 --output-tokens 128 \
  --json gpurun_out/genai_prompt512_c4.json > gpurun_out/peak_ll.log 2>&1
echo "genai rc=$?"
python -c "
import json; d=json.load(open('gpurun_out/genai_prompt512_c4.json'))
print('tok/s',d['output_tokens_per_sec'],'TTFT',d['ttft_ms'],'ITL p50/p99',d['inter_token_latency_ms']['p50'],d['inter_token_latency_ms']['p99'],'err',d['errors'])" || true
kill $SRV 2>/dev/null; wait $SRV 2>/dev/null
echo DONE
