# Prefill take 2: per-group graphs (row_map) — stall should now be a
# [1,256] replay. Targets: long512 max ITL ~ p99 (~12 ms), TTFT p50
# well under 100 ms, tok/s >= 800.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python -c "import __graft_entry__ as g; g.build()" > gpurun_out/r02p2_build.log 2>&1
echo "build rc=$?"

python -m client_amd.server --models llama3_8b --grpc-port 18001 > gpurun_out/r02p2_llama_server.log 2>&1 &
SRV=$!
for i in $(seq 120); do grep -q GRPC_READY gpurun_out/r02p2_llama_server.log && break; sleep 2; done

timeout 600 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 8 --requests 16 --prompt-tokens 512 --output-tokens 512 \
  --json gpurun_out/r02p2_genai_long512_c8.json > gpurun_out/r02p2_genai1.log 2>&1
echo "genai long rc=$?"

timeout 420 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 8 --requests 16 --prompt-tokens 128 --output-tokens 256 \
  --json gpurun_out/r02p2_genai_c8.json > gpurun_out/r02p2_genai2.log 2>&1
echo "genai c8 rc=$?"

for f in gpurun_out/r02p2_genai_long512_c8.json gpurun_out/r02p2_genai_c8.json; do
  python - "$f" <<'PYEOF'
import json, sys
d = json.load(open(sys.argv[1]))
print(sys.argv[1], '-> tok/s', d['output_tokens_per_sec'],
      'ITL p50/p99/max', d['inter_token_latency_ms']['p50'],
      d['inter_token_latency_ms']['p99'], d['inter_token_latency_ms']['max'],
      'TTFT p50/p99', d['ttft_ms']['p50'], d['ttft_ms']['p99'],
      'err', d['errors'])
PYEOF
done
kill $SRV 2>/dev/null
wait $SRV 2>/dev/null
echo DONE
