# Measure TTFT tail after decode-graph prewarm (was p90 590ms cold).
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
python -m client_amd.server --models llama3_8b --grpc-port 18001 > gpurun_out/ttft_server.log 2>&1 &
SRV=$!
for i in $(seq 90); do grep -q GRPC_READY gpurun_out/ttft_server.log && break; sleep 2; done
timeout 300 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 4 --requests 16 --prompt-tokens 128 --output-tokens 128 \
  --json gpurun_out/genai_prewarm_c4.json > gpurun_out/ttft_genai.log 2>&1
echo "genai rc=$?"
cat gpurun_out/genai_prewarm_c4.json
kill $SRV 2>/dev/null
wait $SRV 2>/dev/null
echo DONE
