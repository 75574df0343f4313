#!/usr/bin/env python3
"""Async (futures) HTTP inference
(reference: simple_http_async_infer_client.py)."""
import argparse

import numpy as np

import tritonclient.http as httpclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8000")
    args = parser.parse_args()

    with httpclient.InferenceServerClient(args.url, concurrency=4) as client:
        requests = []
        for i in range(8):
            inputs = [
                httpclient.InferInput("INPUT0", [1, 16], "INT32"),
                httpclient.InferInput("INPUT1", [1, 16], "INT32"),
            ]
            inputs[0].set_data_from_numpy(np.full((1, 16), i, dtype=np.int32))
            inputs[1].set_data_from_numpy(np.ones((1, 16), dtype=np.int32))
            requests.append((i, client.async_infer("simple", inputs)))
        for i, request in requests:
            result = request.get_result()
            out = result.as_numpy("OUTPUT0")
            assert out[0][0] == i + 1
        print("PASS: async infer")
