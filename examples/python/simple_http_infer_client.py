#!/usr/bin/env python3
"""Sync HTTP inference against the 'simple' addsub model
(reference: src/python/examples/simple_http_infer_client.py)."""
import argparse
import sys

import numpy as np

import tritonclient.http as httpclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8000")
    parser.add_argument("-v", "--verbose", action="store_true")
    args = parser.parse_args()

    with httpclient.InferenceServerClient(args.url, verbose=args.verbose) as client:
        inputs = [
            httpclient.InferInput("INPUT0", [1, 16], "INT32"),
            httpclient.InferInput("INPUT1", [1, 16], "INT32"),
        ]
        input0_data = np.arange(16, dtype=np.int32).reshape(1, 16)
        input1_data = np.ones((1, 16), dtype=np.int32)
        inputs[0].set_data_from_numpy(input0_data, binary_data=True)
        inputs[1].set_data_from_numpy(input1_data, binary_data=False)
        outputs = [
            httpclient.InferRequestedOutput("OUTPUT0", binary_data=True),
            httpclient.InferRequestedOutput("OUTPUT1", binary_data=False),
        ]
        results = client.infer("simple", inputs, outputs=outputs)
        output0_data = results.as_numpy("OUTPUT0")
        output1_data = results.as_numpy("OUTPUT1")
        for i in range(16):
            print(f"{input0_data[0][i]} + {input1_data[0][i]} = {output0_data[0][i]}")
            if (input0_data[0][i] + input1_data[0][i]) != output0_data[0][i]:
                sys.exit("addition error")
            if (input0_data[0][i] - input1_data[0][i]) != output1_data[0][i]:
                sys.exit("subtraction error")
        print("PASS: infer")
