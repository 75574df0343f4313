# Mixed-load co-serving re-soak: ResNet + BERT(graphs) + Llama(prewarm)
# on one GPU simultaneously; BERT captures shapes while llama decode
# graphs replay — the capture-poisoning regression scenario.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
python -m client_amd.server --models resnet50,bert_large,llama3_8b \
  --grpc-port 18004 --dynamic-batching > gpurun_out/r02soak_server.log 2>&1 &
SRV=$!
for i in $(seq 120); do grep -q GRPC_READY gpurun_out/r02soak_server.log && break; sleep 2; done

timeout 240 python -m client_amd.perf -m resnet50 -u 127.0.0.1:18004 -i grpc -b 8 \
  --shared-memory cuda --concurrency-range 4:4:1 --measurement-interval 2 \
  --warmup 2 --max-windows 6 --json gpurun_out/r02soak_resnet.json > gpurun_out/r02soak_resnet.log 2>&1 &
P1=$!
timeout 240 python -m client_amd.perf -m bert_large -u 127.0.0.1:18004 -i grpc -b 8 \
  --shared-memory cuda --shape input_ids:128 --concurrency-range 4:4:1 \
  --measurement-interval 2 --warmup 2 --max-windows 6 \
  --json gpurun_out/r02soak_bert.json > gpurun_out/r02soak_bert.log 2>&1 &
P2=$!
timeout 240 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18004 \
  --concurrency 4 --requests 10 --prompt-tokens 128 --output-tokens 96 \
  --json gpurun_out/r02soak_llama.json > gpurun_out/r02soak_llama.log 2>&1 &
P3=$!
wait $P1; echo "resnet rc=$?"
wait $P2; echo "bert rc=$?"
wait $P3; echo "llama rc=$?"
for f in r02soak_resnet r02soak_bert r02soak_llama; do
  echo "== $f"; python -c "
import json
d=json.load(open('gpurun_out/$f.json'))
d=d[0] if isinstance(d,list) else d
print({k:d[k] for k in d if 'err' in k or 'per_sec' in k or k=='inferences_per_sec'})" || true
done
kill $SRV 2>/dev/null; wait $SRV 2>/dev/null
echo DONE
# plus a long-generation stability leg: 8 x 700-token streams
python -m client_amd.server --models llama3_8b --grpc-port 18008 > gpurun_out/r02soak_ll700.log 2>&1 &
L=$!
for i in $(seq 150); do grep -q GRPC_READY gpurun_out/r02soak_ll700.log && break; sleep 2; done
timeout 600 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18008 \
  --concurrency 8 --requests 8 --prompt-tokens 128 --output-tokens 700 \
  --json gpurun_out/r02soak_long700.json > gpurun_out/r02soak_long700_cli.log 2>&1
echo "long700 rc=$?"
python - gpurun_out/r02soak_long700.json <<'PYEOF'
import json, sys
d = json.load(open(sys.argv[1]))
print('long700 -> tok/s', d['output_tokens_per_sec'],
      'ITL p50/p99/max', d['inter_token_latency_ms']['p50'],
      d['inter_token_latency_ms']['p99'], d['inter_token_latency_ms']['max'], 'err', d['errors'])
PYEOF
kill $L 2>/dev/null; wait $L 2>/dev/null
echo DONE2
