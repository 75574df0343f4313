# Mixed-load co-serving re-soak: ResNet + BERT(graphs) + Llama(prewarm)
# on one GPU simultaneously; BERT captures shapes while llama decode
# graphs replay — the capture-poisoning regression scenario.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
python -m client_amd.server --models resnet50,bert_large,llama3_8b \
  --grpc-port 18004 --dynamic-batching > gpurun_out/soak2_server.log 2>&1 &
SRV=$!
for i in $(seq 120); do grep -q GRPC_READY gpurun_out/soak2_server.log && break; sleep 2; done

timeout 240 python -m client_amd.perf -m resnet50 -u 127.0.0.1:18004 -i grpc -b 8 \
  --shared-memory cuda --concurrency-range 4:4:1 --measurement-interval 2 \
  --warmup 2 --max-windows 6 --json gpurun_out/soak2_resnet.json > gpurun_out/soak2_resnet.log 2>&1 &
P1=$!
timeout 240 python -m client_amd.perf -m bert_large -u 127.0.0.1:18004 -i grpc -b 8 \
  --shared-memory cuda --shape input_ids:128 --concurrency-range 4:4:1 \
  --measurement-interval 2 --warmup 2 --max-windows 6 \
  --json gpurun_out/soak2_bert.json > gpurun_out/soak2_bert.log 2>&1 &
P2=$!
timeout 240 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18004 \
  --concurrency 4 --requests 10 --prompt-tokens 128 --output-tokens 96 \
  --json gpurun_out/soak2_llama.json > gpurun_out/soak2_llama.log 2>&1 &
P3=$!
wait $P1; echo "resnet rc=$?"
wait $P2; echo "bert rc=$?"
wait $P3; echo "llama rc=$?"
for f in soak2_resnet soak2_bert soak2_llama; do
  echo "== $f"; python -c "
import json
d=json.load(open('gpurun_out/$f.json'))
d=d[0] if isinstance(d,list) else d
print({k:d[k] for k in d if 'err' in k or 'per_sec' in k or k=='inferences_per_sec'})" || true
done
kill $SRV 2>/dev/null; wait $SRV 2>/dev/null
echo DONE
