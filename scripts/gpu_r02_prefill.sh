# Round-2 prefill validation: captured batched prefill must kill the
# ~86 ms ITL stall (target: max ITL ~ p99 ~ 11 ms) and cut TTFT.
# Also re-runs the two fixed kernel tests.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python -c "import __graft_entry__ as g; g.build()" > gpurun_out/r02p_build.log 2>&1
echo "build rc=$?"

timeout 600 python -m pytest tests/test_hip_shm_gpu.py -q > gpurun_out/r02p_pytest.log 2>&1
echo "pytest rc=$?"
grep -E "passed|failed" gpurun_out/r02p_pytest.log

python -m client_amd.server --models llama3_8b --grpc-port 18001 > gpurun_out/r02p_llama_server.log 2>&1 &
SRV=$!
for i in $(seq 120); do grep -q GRPC_READY gpurun_out/r02p_llama_server.log && break; sleep 2; done

# the r01 comparison soak: 8 streams x 512-token prompts, 512 new tokens
timeout 600 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 8 --requests 16 --prompt-tokens 512 --output-tokens 512 \
  --json gpurun_out/r02p_genai_long512_c8.json > gpurun_out/r02p_genai1.log 2>&1
echo "genai long rc=$?"
cat gpurun_out/r02p_genai_long512_c8.json 2>/dev/null | python -c "import json,sys; d=json.load(sys.stdin); print('tok/s', d['output_tokens_per_sec'], 'ITL p50/p99/max', d['inter_token_latency_ms']['p50'], d['inter_token_latency_ms']['p99'], d['inter_token_latency_ms']['max'], 'TTFT p50/p99', d['ttft_ms']['p50'], d['ttft_ms']['p99'])"

# standard c8 comparison point (128-token prompts)
timeout 420 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 8 --requests 16 --prompt-tokens 128 --output-tokens 256 \
  --json gpurun_out/r02p_genai_c8.json > gpurun_out/r02p_genai2.log 2>&1
echo "genai c8 rc=$?"
cat gpurun_out/r02p_genai_c8.json 2>/dev/null | python -c "import json,sys; d=json.load(sys.stdin); print('tok/s', d['output_tokens_per_sec'], 'ITL p50/p99/max', d['inter_token_latency_ms']['p50'], d['inter_token_latency_ms']['p99'], d['inter_token_latency_ms']['max'], 'TTFT p50/p99', d['ttft_ms']['p50'], d['ttft_ms']['p99'])"

kill $SRV 2>/dev/null
wait $SRV 2>/dev/null
echo DONE
