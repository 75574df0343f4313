set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python -c "import __graft_entry__ as g; g.build(); g.smoke(); print('SMOKE-OK')" > gpurun_out/f3_entry.log 2>&1
echo "entry rc=$?"; tail -2 gpurun_out/f3_entry.log

timeout 1000 python -m pytest tests -m gpu -q > gpurun_out/f3_pytest.log 2>&1
echo "pytest rc=$?"; grep -E "passed|failed" gpurun_out/f3_pytest.log | tail -1

timeout 300 python bench.py --steps 10 --warmup 3 > gpurun_out/f3_bench.log 2>&1
echo "bench rc=$?"; tail -1 gpurun_out/f3_bench.log | head -c 260; echo

# --model-warmup smoke: server pre-captures, then one cold client infer
python -m client_amd.server --models resnet50 --grpc-port 18003 --dynamic-batching --model-warmup > gpurun_out/f3_warm_server.log 2>&1 &
WS=$!
for i in $(seq 90); do grep -q GRPC_READY gpurun_out/f3_warm_server.log && break; sleep 2; done
timeout 120 python - <<'PY' > gpurun_out/f3_warm_check.log 2>&1
import numpy as np, time
import client_amd.grpc as g
c = g.InferenceServerClient("127.0.0.1:18003")
x = np.random.rand(8,3,224,224).astype(np.float32)
inp = g.InferInput("INPUT0", [8,3,224,224], "BF16"); inp.set_data_from_numpy(x)
t0=time.monotonic(); r = c.infer("resnet50", [inp]); dt=(time.monotonic()-t0)*1e3
print("cold-infer-ms", round(dt,1), "out", r.as_numpy("OUTPUT0").shape)
c.close()
PY
echo "warmcheck rc=$?"; cat gpurun_out/f3_warm_check.log
kill $WS 2>/dev/null; wait $WS 2>/dev/null

# fresh llama c8 headline with prewarm
python -m client_amd.server --models llama3_8b --grpc-port 18001 > gpurun_out/f3_llama_server.log 2>&1 &
SRV=$!
for i in $(seq 90); do grep -q GRPC_READY gpurun_out/f3_llama_server.log && break; sleep 2; done
timeout 300 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 8 --requests 24 --prompt-tokens 128 --output-tokens 128 \
  --json gpurun_out/genai_prewarm_c8.json > gpurun_out/f3_genai.log 2>&1
echo "genai rc=$?"; cat gpurun_out/genai_prewarm_c8.json 2>/dev/null
kill $SRV 2>/dev/null; wait $SRV 2>/dev/null
echo DONE
