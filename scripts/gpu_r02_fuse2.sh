# RoPE-scatter fusion + mask-cast hoist validation: numerics + genai.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python -c "import __graft_entry__ as g; g.build()" > gpurun_out/r02f_build.log 2>&1
echo "build rc=$?"

timeout 600 python -m pytest tests/test_hip_shm_gpu.py -q > gpurun_out/r02f_pytest.log 2>&1
echo "pytest rc=$?"; grep -E "passed|failed" gpurun_out/r02f_pytest.log

python -m client_amd.server --models llama3_8b --grpc-port 18001 > gpurun_out/r02f_ll.log 2>&1 &
SRV=$!
for i in $(seq 150); do grep -q GRPC_READY gpurun_out/r02f_ll.log && break; sleep 2; done
timeout 420 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 8 --requests 16 --prompt-tokens 128 --output-tokens 256 \
  --json gpurun_out/r02f_genai_c8.json > gpurun_out/r02f_g1.log 2>&1
echo "genai c8 rc=$?"
timeout 600 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 8 --requests 12 --prompt-tokens 512 --output-tokens 512 \
  --json gpurun_out/r02f_genai_long.json > gpurun_out/r02f_g2.log 2>&1
echo "genai long rc=$?"
for f in gpurun_out/r02f_genai_c8.json gpurun_out/r02f_genai_long.json; do
  python - "$f" <<'PYEOF'
import json, sys
d = json.load(open(sys.argv[1]))
print(sys.argv[1].split('/')[-1], '-> tok/s', d['output_tokens_per_sec'],
      'ITL p50/p99/max', d['inter_token_latency_ms']['p50'],
      d['inter_token_latency_ms']['p99'], d['inter_token_latency_ms']['max'],
      'TTFT p50/p99', d['ttft_ms']['p50'], d['ttft_ms']['p99'], 'err', d['errors'])
PYEOF
done
kill $SRV 2>/dev/null; wait $SRV 2>/dev/null
echo DONE
