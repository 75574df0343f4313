# Validation 6: event-reuse fix (DenseNet/ResNet recovery) with a
# device-sync A/B, plus GQA-bmm decode A/B.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python -c "import __graft_entry__ as g; g.build()" > gpurun_out/r02v6_build.log 2>&1
echo "build rc=$?"

run_dn () {  # $1 = tag, $2 = env assignment
  env $2 python -m client_amd.server --models densenet121 --grpc-port 18005 --dynamic-batching > gpurun_out/r02v6_dn_server_$1.log 2>&1 &
  SPID=$!
  for i in $(seq 90); do grep -q GRPC_READY gpurun_out/r02v6_dn_server_$1.log && break; sleep 2; done
  timeout 240 python -m client_amd.perf -m densenet121 -u 127.0.0.1:18005 -i grpc -b 8 \
    --shared-memory cuda --concurrency-range 8:8:1 --measurement-interval 2 \
    --warmup 4 --max-windows 3 --json gpurun_out/r02v6_densenet_$1.json > gpurun_out/r02v6_dn_$1.log 2>&1
  echo "densenet $1 rc=$?"
  python -c "
import json
for r in json.load(open('gpurun_out/r02v6_densenet_$1.json')): print('$1 c',r['concurrency'],r['inferences_per_sec'],'p99us',r['latency_us']['p99'],'server',r.get('server'))" || true
  kill $SPID 2>/dev/null; wait $SPID 2>/dev/null
}

run_dn evreuse "CLIENT_AMD_SYNC_MODE=event"
run_dn devsync "CLIENT_AMD_SYNC_MODE=device"

timeout 300 python bench.py --steps 15 --warmup 5 > gpurun_out/r02v6_bench.log 2>&1
echo "bench rc=$?"; tail -1 gpurun_out/r02v6_bench.log | head -c 260; echo

run_llama () {  # $1 = tag, $2 = env
  env $2 CLIENT_AMD_DECODE_TRACE=1 python -m client_amd.server --models llama3_8b --grpc-port 18001 > gpurun_out/r02v6_ll_server_$1.log 2>&1 &
  LPID=$!
  for i in $(seq 120); do grep -q GRPC_READY gpurun_out/r02v6_ll_server_$1.log && break; sleep 2; done
  timeout 420 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
    --concurrency 8 --requests 12 --prompt-tokens 128 --output-tokens 256 \
    --json gpurun_out/r02v6_genai_$1.json > gpurun_out/r02v6_genai_$1.log 2>&1
  echo "genai $1 rc=$?"
  grep "decode-trace" gpurun_out/r02v6_ll_server_$1.log | tail -3
  python - gpurun_out/r02v6_genai_$1.json <<'PYEOF'
import json, sys
d = json.load(open(sys.argv[1]))
print(sys.argv[1].split('/')[-1], '-> tok/s', d['output_tokens_per_sec'],
      'ITL p50/p99/max', d['inter_token_latency_ms']['p50'],
      d['inter_token_latency_ms']['p99'], d['inter_token_latency_ms']['max'],
      'err', d['errors'])
PYEOF
  kill $LPID 2>/dev/null; wait $LPID 2>/dev/null
}

run_llama bmm "CLIENT_AMD_GQA_BMM=1"
run_llama sdpa "CLIENT_AMD_GQA_BMM=0"
echo DONE
