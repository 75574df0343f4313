# Round-2 sanity: correctness fixes (KV scratch slot, shm bounds,
# serializer return type) + rebuilt extension with the new fan-out
# bindings. build+smoke, gpu pytest, short bench, short genai.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python -c "import __graft_entry__ as g; g.build(); g.smoke(); print('SMOKE-OK')" > gpurun_out/r02_entry.log 2>&1
echo "entry rc=$?"
tail -2 gpurun_out/r02_entry.log

timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r02_pytest.log 2>&1
echo "pytest rc=$?"
tail -3 gpurun_out/r02_pytest.log

timeout 300 python bench.py --steps 15 --warmup 5 > gpurun_out/r02_bench.log 2>&1
echo "bench rc=$?"
tail -1 gpurun_out/r02_bench.log

python -m client_amd.server --models llama3_8b --grpc-port 18001 > gpurun_out/r02_llama_server.log 2>&1 &
SRV=$!
for i in $(seq 90); do grep -q GRPC_READY gpurun_out/r02_llama_server.log && break; sleep 2; done
timeout 360 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
  --concurrency 8 --requests 16 --prompt-tokens 128 --output-tokens 256 \
  --json gpurun_out/r02_genai_c8.json > gpurun_out/r02_genai.log 2>&1
echo "genai rc=$?"
cat gpurun_out/r02_genai_c8.json 2>/dev/null
kill $SRV 2>/dev/null
wait $SRV 2>/dev/null
echo DONE
