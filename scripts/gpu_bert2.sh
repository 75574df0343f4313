set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
python -m client_amd.server --models bert_large --grpc-port 18002 --dynamic-batching > gpurun_out/bert2_server.log 2>&1 &
SRV=$!
for i in $(seq 90); do grep -q GRPC_READY gpurun_out/bert2_server.log && break; sleep 2; done
timeout 420 python -m client_amd.perf -m bert_large -u 127.0.0.1:18002 -i grpc -b 8 \
  --shared-memory cuda --shape input_ids:128 --concurrency-range 8:8:1 \
  --measurement-interval 2 --warmup 8 --max-windows 6 \
  --json gpurun_out/bert2_c8.json > gpurun_out/bert2_perf.log 2>&1
echo "perf rc=$?"
cat gpurun_out/bert2_c8.json 2>/dev/null
kill $SRV 2>/dev/null; wait $SRV 2>/dev/null
echo DONE
