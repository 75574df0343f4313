"""genai-perf-class LLM load generator.

The reference relocates genai-perf out of its snapshot
(src/c++/perf_analyzer/genai-perf/README.md); this is the equivalent
for this stack: drives N concurrent decoupled gRPC token streams
against a GenerateModel and reports LLM serving metrics — time to
first token (TTFT), inter-token latency (ITL), and aggregate output
token throughput.

    python -m client_amd.perf.genai -m llama3_8b -u 127.0.0.1:8001 \
        --concurrency 4 --prompt-tokens 128 --output-tokens 64 --requests 8
"""

import argparse
import json
import queue
import threading
import time

import numpy as np


def percentile(sorted_vals, q):
    if not sorted_vals:
        return 0.0
    idx = min(len(sorted_vals) - 1,
              int(round(q / 100.0 * (len(sorted_vals) - 1))))
    return sorted_vals[idx]


def _stats_ms(sorted_vals, scale=1000.0, digits=3):
    if not sorted_vals:
        return {"avg": 0, "min": 0, "max": 0, "p50": 0, "p90": 0, "p99": 0}
    return {
        "avg": round(sum(sorted_vals) / len(sorted_vals) * scale, digits),
        "min": round(sorted_vals[0] * scale, digits),
        "max": round(sorted_vals[-1] * scale, digits),
        "p50": round(percentile(sorted_vals, 50) * scale, digits),
        "p90": round(percentile(sorted_vals, 90) * scale, digits),
        "p99": round(percentile(sorted_vals, 99) * scale, digits),
    }


class GenAiPerf:
    def __init__(self, url, model_name, prompt_tokens=128, output_tokens=64,
                 vocab_size=128256, verbose=False, warmup_requests=0,
                 seed=None):
        self.url = url
        self.model_name = model_name
        self.prompt_tokens = prompt_tokens
        self.output_tokens = output_tokens
        self.vocab_size = vocab_size
        self.verbose = verbose
        # per-stream requests issued before measurement (genai-perf
        # --warmup-request-count)
        self.warmup_requests = warmup_requests
        self.seed = seed

    def _one_stream_worker(self, n_requests, out):
        import client_amd.grpc as grpcclient

        rng = np.random.default_rng(self.seed)
        client = grpcclient.InferenceServerClient(self.url)
        events = queue.Queue()
        client.start_stream(
            callback=lambda result, error: events.put((result, error)))
        try:
            for req_i in range(self.warmup_requests + n_requests):
                warm = req_i < self.warmup_requests
                ids = rng.integers(
                    0, self.vocab_size, self.prompt_tokens
                ).astype(np.int64)
                inputs = [
                    grpcclient.InferInput(
                        "input_ids", [self.prompt_tokens], "INT64"),
                    grpcclient.InferInput("max_tokens", [1], "INT32"),
                ]
                inputs[0].set_data_from_numpy(ids)
                inputs[1].set_data_from_numpy(
                    np.array([self.output_tokens], dtype=np.int32))
                t0 = time.monotonic()
                client.async_stream_infer(
                    self.model_name, inputs,
                    enable_empty_final_response=True)
                stamps = []
                error = None
                while True:
                    result, err = events.get(timeout=600)
                    if err is not None:
                        error = err
                        break
                    if result.is_final_response():
                        break
                    stamps.append(time.monotonic())
                if error is not None:
                    if not warm:
                        out["errors"].append(str(error))
                    continue
                if stamps and not warm:
                    out["ttft"].append(stamps[0] - t0)
                    out["latency"].append(stamps[-1] - t0)
                    out["tokens"].append(len(stamps))
                    for a, b in zip(stamps, stamps[1:]):
                        out["itl"].append(b - a)
        finally:
            client.stop_stream()
            client.close()

    def run(self, concurrency=1, requests_per_stream=4):
        out = {"ttft": [], "itl": [], "tokens": [], "errors": [],
               "latency": []}
        threads = []
        t_start = time.monotonic()
        for _ in range(concurrency):
            t = threading.Thread(
                target=self._one_stream_worker, args=(requests_per_stream, out))
            t.start()
            threads.append(t)
        for t in threads:
            t.join()
        elapsed = time.monotonic() - t_start
        ttft = sorted(out["ttft"])
        itl = sorted(out["itl"])
        latency = sorted(out["latency"])
        total_tokens = sum(out["tokens"])
        n_req = concurrency * requests_per_stream
        return {
            "model": self.model_name,
            "concurrency": concurrency,
            "requests": n_req,
            "warmup_requests_per_stream": self.warmup_requests,
            "prompt_tokens": self.prompt_tokens,
            "output_tokens_per_request": self.output_tokens,
            "total_output_tokens": total_tokens,
            "output_tokens_per_sec": round(total_tokens / elapsed, 2),
            "request_throughput_per_sec": round(n_req / elapsed, 3),
            "ttft_ms": _stats_ms(ttft, digits=2),
            "inter_token_latency_ms": _stats_ms(itl),
            "request_latency_ms": _stats_ms(latency, digits=2),
            "errors": len(out["errors"]),
            "elapsed_s": round(elapsed, 2),
        }


def main(argv=None):
    p = argparse.ArgumentParser("client_amd.perf.genai")
    p.add_argument("-m", "--model-name", required=True)
    p.add_argument("-u", "--url", default="127.0.0.1:8001")
    p.add_argument("--concurrency", type=int, default=1)
    p.add_argument("--requests", type=int, default=4,
                   help="requests per stream")
    p.add_argument("--prompt-tokens", type=int, default=128)
    p.add_argument("--output-tokens", type=int, default=64)
    p.add_argument("--vocab-size", type=int, default=128256)
    p.add_argument("--warmup-request-count", type=int, default=0,
                   help="per-stream unmeasured warmup requests")
    p.add_argument("--random-seed", type=int, default=None)
    p.add_argument("--json", default=None)
    args = p.parse_args(argv)

    ga = GenAiPerf(
        url=args.url, model_name=args.model_name,
        prompt_tokens=args.prompt_tokens, output_tokens=args.output_tokens,
        vocab_size=args.vocab_size,
        warmup_requests=args.warmup_request_count, seed=args.random_seed,
    )
    result = ga.run(concurrency=args.concurrency,
                    requests_per_stream=args.requests)
    print(json.dumps(result, indent=2))
    if args.json:
        with open(args.json, "w") as f:
            json.dump(result, f, indent=2)


if __name__ == "__main__":
    main()
