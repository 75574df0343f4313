#!/usr/bin/env python3
"""Repeated-inference RSS watch for leak detection
(reference: memory_growth_test.py + C++ memory_leak_test.cc)."""
import argparse
import resource

import numpy as np

import tritonclient.http as httpclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8000")
    parser.add_argument("-n", "--iterations", type=int, default=500)
    parser.add_argument("--max-growth-mb", type=float, default=64.0)
    args = parser.parse_args()

    with httpclient.InferenceServerClient(args.url) as client:
        x = np.random.rand(1, 1024).astype(np.float32)
        inp = httpclient.InferInput("INPUT0", [1, 1024], "FP32")

        def one():
            inp.set_data_from_numpy(x)
            result = client.infer("identity_fp32", [inp])
            assert result.as_numpy("OUTPUT0") is not None

        for _ in range(20):
            one()  # warmup
        rss0 = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
        for _ in range(args.iterations):
            one()
        rss1 = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
        growth_mb = (rss1 - rss0) / 1024.0
        print(f"RSS growth over {args.iterations} inferences: {growth_mb:.1f} MB")
        assert growth_mb < args.max_growth_mb, "memory growth detected"
        print("PASS: memory growth")
