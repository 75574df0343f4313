"""Decoupled LLM token-streaming benchmark.

Spawns a server with the Llama-3-8B-class model (random-init bf16) and
streams tokens over the decoupled gRPC path; reports prefill latency
and steady-state decode tokens/s.

    python scripts/bench_llama.py [--model llama3_8b] [--prompt 128]
        [--new-tokens 64]
"""

import argparse
import os
import subprocess
import sys
import time
import queue

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3_8b",
                    choices=["llama3_8b", "llama_tiny"])
    ap.add_argument("--prompt", type=int, default=128)
    ap.add_argument("--new-tokens", type=int, default=64)
    args = ap.parse_args()

    import client_amd.grpc as grpcclient

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.Popen(
        [sys.executable, "-m", "client_amd.server", "--grpc-port", "-1",
         "--models", args.model, "--device", "cuda:0", "--dtype", "bf16"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True, cwd=repo,
    )
    try:
        port = None
        deadline = time.time() + 600
        while time.time() < deadline:
            line = proc.stdout.readline()
            if not line:
                if proc.poll() is not None:
                    raise SystemExit("server died during init")
                time.sleep(0.1)
                continue
            sys.stderr.write(line)
            if line.startswith("GRPC_READY"):
                port = int(line.split()[1])
                break
        assert port, "server not ready"
        client = grpcclient.InferenceServerClient(f"127.0.0.1:{port}")
        results = queue.Queue()
        client.start_stream(
            callback=lambda result, error: results.put((result, error)),
            stream_timeout=None,
        )
        vocab = 128256 if args.model == "llama3_8b" else 256
        ids = np.random.randint(0, vocab, args.prompt).astype(np.int64)
        inputs = [
            grpcclient.InferInput("input_ids", [args.prompt], "INT64"),
            grpcclient.InferInput("max_tokens", [1], "INT32"),
        ]
        inputs[0].set_data_from_numpy(ids)
        inputs[1].set_data_from_numpy(
            np.array([args.new_tokens], dtype=np.int32))
        t0 = time.monotonic()
        client.async_stream_infer(
            args.model, inputs, enable_empty_final_response=True)
        stamps = []
        while True:
            result, error = results.get(timeout=600)
            if error is not None:
                raise SystemExit(f"stream error: {error}")
            if result.is_final_response():
                break
            stamps.append(time.monotonic())
        client.stop_stream()
        client.close()
        n = len(stamps)
        ttft = stamps[0] - t0
        decode_s = stamps[-1] - stamps[0] if n > 1 else 0.0
        tps = (n - 1) / decode_s if decode_s > 0 else 0.0
        import json

        print(json.dumps({
            "model": args.model,
            "prompt_tokens": args.prompt,
            "new_tokens": n,
            "time_to_first_token_s": round(ttft, 3),
            "decode_tokens_per_sec": round(tps, 2),
            "dtype": "bf16",
            "transport": "grpc decoupled stream",
        }))
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()


if __name__ == "__main__":
    main()
