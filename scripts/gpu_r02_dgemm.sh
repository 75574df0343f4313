# Decode-GEMM validation + A/B: numerics tests, kernel-only bandwidth,
# then genai on/off.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 420 python -c "import __graft_entry__ as g; g.build()" > gpurun_out/r02g_build.log 2>&1
echo "build rc=$?"

timeout 420 python -m pytest tests/test_hip_shm_gpu.py -q -k "decode_gemm" > gpurun_out/r02g_pytest.log 2>&1
echo "pytest rc=$?"; grep -E "passed|failed" gpurun_out/r02g_pytest.log

# kernel-only bandwidth on the big decode shapes
timeout 300 python - > gpurun_out/r02g_bw.log 2>&1 <<'PYEOF'
import time
import torch
from client_amd.ops import hip_runtime as hr
for (n, k) in [(4096, 4096), (14336, 4096), (4096, 14336), (128256, 4096)]:
    x = torch.randn(8, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    y = torch.empty(8, n, device="cuda", dtype=torch.bfloat16)
    s = torch.cuda.current_stream().cuda_stream
    for _ in range(3):
        hr.decode_gemm_bf16(x.data_ptr(), w.data_ptr(), y.data_ptr(), n, k, s)
    torch.cuda.synchronize()
    t0 = time.perf_counter(); iters = 50
    for _ in range(iters):
        hr.decode_gemm_bf16(x.data_ptr(), w.data_ptr(), y.data_ptr(), n, k, s)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    gbs = n * k * 2 / dt / 1e9
    # torch reference timing
    xl = x.clone(); wl = torch.nn.Parameter(w.clone(), requires_grad=False)
    lin = lambda: torch.nn.functional.linear(xl, wl)
    for _ in range(3): lin()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): lin()
    torch.cuda.synchronize()
    dt2 = (time.perf_counter() - t0) / iters
    gbs2 = n * k * 2 / dt2 / 1e9
    print(f"N={n} K={k}: kernel {dt*1e6:.1f}us {gbs:.0f} GB/s | torch {dt2*1e6:.1f}us {gbs2:.0f} GB/s")
PYEOF
echo "bw rc=$?"; cat gpurun_out/r02g_bw.log

run_llama () {  # $1 tag, $2 env
  env $2 python -m client_amd.server --models llama3_8b --grpc-port 18001 > gpurun_out/r02g_ll_$1.log 2>&1 &
  LPID=$!
  for i in $(seq 150); do grep -q GRPC_READY gpurun_out/r02g_ll_$1.log && break; sleep 2; done
  timeout 420 python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:18001 \
    --concurrency 8 --requests 12 --prompt-tokens 128 --output-tokens 256 \
    --json gpurun_out/r02g_genai_$1.json > gpurun_out/r02g_g_$1.log 2>&1
  echo "genai $1 rc=$?"
  python - gpurun_out/r02g_genai_$1.json <<'PYEOF'
import json, sys
d = json.load(open(sys.argv[1]))
print(sys.argv[1].split('/')[-1], '-> tok/s', d['output_tokens_per_sec'],
      'ITL p50/p99/max', d['inter_token_latency_ms']['p50'],
      d['inter_token_latency_ms']['p99'], d['inter_token_latency_ms']['max'], 'err', d['errors'])
PYEOF
  kill $LPID 2>/dev/null; wait $LPID 2>/dev/null
}
run_llama dg "CLIENT_AMD_DECODE_GEMM=1"
run_llama nodg ""
echo DONE
