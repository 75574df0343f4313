# TunableOp A/B on the decode GEMMs (skinny batch-8 GEMMs measured at
# 2.1-4.2 TB/s; tuning targets the ~4.7 ms/step GEMM time) + adaptive
# prefill-cap TTFT check.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python -c "import __graft_entry__ as g; g.build()" > gpurun_out/r02t_build.log 2>&1
echo "build rc=$?"

run_llama () {  # $1 tag, $2 env
  env $2 python -m client_amd.server --models llama3_8b --grpc-port 18001 > gpurun_out/r02t_ll_$1.log 2>&1 &
  LPID=$!
  for i in $(seq 240); do grep -q GRPC_READY gpurun_out/r02t_ll_$1.log && break; sleep 2; done
  grep -q GRPC_READY gpurun_out/r02t_ll_$1.log || echo "$1 server NOT ready"
  timeout 420 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
    --concurrency 8 --requests 12 --prompt-tokens 128 --output-tokens 256 \
    --json gpurun_out/r02t_genai_$1.json > gpurun_out/r02t_g_$1.log 2>&1
  echo "genai $1 rc=$?"
  python - gpurun_out/r02t_genai_$1.json <<'PYEOF'
import json, sys
d = json.load(open(sys.argv[1]))
print(sys.argv[1].split('/')[-1], '-> tok/s', d['output_tokens_per_sec'],
      'ITL p50/p99/max', d['inter_token_latency_ms']['p50'],
      d['inter_token_latency_ms']['p99'], d['inter_token_latency_ms']['max'],
      'TTFT p50/p99', d['ttft_ms']['p50'], d['ttft_ms']['p99'], 'err', d['errors'])
PYEOF
  if [ "$1" = "tuned" ]; then
    # also the long-prompt TTFT check with the adaptive cap
    timeout 600 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
      --concurrency 8 --requests 12 --prompt-tokens 512 --output-tokens 512 \
      --json gpurun_out/r02t_genai_long_$1.json > gpurun_out/r02t_gl_$1.log 2>&1
    python - gpurun_out/r02t_genai_long_$1.json <<'PYEOF'
import json, sys
d = json.load(open(sys.argv[1]))
print('long512', '-> tok/s', d['output_tokens_per_sec'],
      'ITL p50/p99/max', d['inter_token_latency_ms']['p50'],
      d['inter_token_latency_ms']['p99'], d['inter_token_latency_ms']['max'],
      'TTFT p50/p99', d['ttft_ms']['p50'], d['ttft_ms']['p99'], 'err', d['errors'])
PYEOF
  fi
  kill $LPID 2>/dev/null; wait $LPID 2>/dev/null
}

run_llama base ""
run_llama tuned "PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 PYTORCH_TUNABLEOP_FILENAME=/root/repo/gpurun_out/tunableop_llama.csv"
head -20 gpurun_out/tunableop_llama* 2>/dev/null
echo DONE
