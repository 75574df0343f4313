# Round-2 final driver mirror: build + smoke + full GPU pytest +
# bench at driver-like settings + resnet sweep for the record.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 480 python -c "import __graft_entry__ as g; g.build(); g.smoke(); print('SMOKE-OK')" > gpurun_out/r02m_entry.log 2>&1
echo "entry rc=$?"; tail -2 gpurun_out/r02m_entry.log

timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r02m_pytest.log 2>&1
echo "pytest rc=$?"; grep -E "passed|failed" gpurun_out/r02m_pytest.log

timeout 420 python bench.py --steps 30 --warmup 8 > gpurun_out/r02m_bench.log 2>&1
echo "bench rc=$?"; tail -1 gpurun_out/r02m_bench.log | head -c 400; echo

python -m client_amd.server --models resnet50 --grpc-port 18007 --dynamic-batching --preferred-batch-size 32 --max-queue-delay-us 400 > gpurun_out/r02m_rs_server.log 2>&1 &
R=$!
for i in $(seq 90); do grep -q GRPC_READY gpurun_out/r02m_rs_server.log && break; sleep 2; done
timeout 300 python -m client_amd.perf -m resnet50 -u 127.0.0.1:18007 -i grpc -b 8 \
  --shared-memory cuda --concurrency-range 1:8:1 --measurement-interval 2 \
  --warmup 3 --max-windows 3 --json gpurun_out/r02m_resnet_sweep.json > gpurun_out/r02m_rs.log 2>&1
echo "sweep rc=$?"
python -c "
import json
for r in json.load(open('gpurun_out/r02m_resnet_sweep.json')): print('resnet c',r['concurrency'],r['inferences_per_sec'],'p99us',r['latency_us']['p99'])" || true
kill $R 2>/dev/null; wait $R 2>/dev/null
echo DONE
