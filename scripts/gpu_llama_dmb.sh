# Llama decode-max-batch sweep: aggregate tok/s + ITL at c8/c16 with
# scheduler batch 8 vs 16.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

run_one() {  # $1=decode_max_batch $2=concurrency $3=requests
  python -m client_amd.server --models llama3_8b --grpc-port 18001 \
    --decode-max-batch $1 > gpurun_out/dmb_server_$1_$2.log 2>&1 &
  SRV=$!
  for i in $(seq 90); do
    grep -q GRPC_READY gpurun_out/dmb_server_$1_$2.log && break; sleep 2
  done
  timeout 300 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
    --concurrency $2 --requests $3 --prompt-tokens 128 --output-tokens 128 \
    --json gpurun_out/genai_dmb$1_c$2.json > gpurun_out/genai_dmb$1_c$2.log 2>&1
  echo "dmb=$1 c=$2 rc=$?"
  cat gpurun_out/genai_dmb$1_c$2.json 2>/dev/null | python -c "import json,sys; d=json.load(sys.stdin); print('tok/s', d['output_tokens_per_sec'], 'ITL p50', d['inter_token_latency_ms']['p50'], 'errors', d['errors'])" || true
  kill $SRV 2>/dev/null
  wait $SRV 2>/dev/null
}

run_one 8 8 16
run_one 16 16 32
run_one 16 8 16
echo DONE
