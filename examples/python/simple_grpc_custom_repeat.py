#!/usr/bin/env python3
"""Decoupled model: N responses per request with final-response flag
(reference: simple_grpc_custom_repeat.cc:135-176 / repeat_int32)."""
import argparse
import queue

import numpy as np

import tritonclient.grpc as grpcclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    parser.add_argument("-n", "--repeat-count", type=int, default=5)
    args = parser.parse_args()

    with grpcclient.InferenceServerClient(args.url) as client:
        results = queue.Queue()
        client.start_stream(callback=lambda result, error: results.put((result, error)))
        n = args.repeat_count
        inputs = [
            grpcclient.InferInput("IN", [n], "INT32"),
            grpcclient.InferInput("DELAY", [n], "UINT32"),
            grpcclient.InferInput("WAIT", [1], "UINT32"),
        ]
        inputs[0].set_data_from_numpy(np.arange(n, dtype=np.int32) * 10)
        inputs[1].set_data_from_numpy(np.zeros(n, dtype=np.uint32))
        inputs[2].set_data_from_numpy(np.zeros(1, dtype=np.uint32))
        client.async_stream_infer(
            "repeat_int32", inputs, enable_empty_final_response=True)
        seen = []
        while True:
            result, error = results.get(timeout=30)
            assert error is None
            if result.is_final_response():
                break
            seen.append(int(result.as_numpy("OUT")[0]))
        assert seen == [i * 10 for i in range(n)]
        client.stop_stream()
        print("PASS: decoupled repeat")
