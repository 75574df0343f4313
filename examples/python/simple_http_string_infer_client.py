#!/usr/bin/env python3
"""BYTES/string tensors over HTTP (reference: simple_http_string_infer_client.py)."""
import argparse

import numpy as np

import tritonclient.http as httpclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8000")
    args = parser.parse_args()

    with httpclient.InferenceServerClient(args.url) as client:
        in0 = np.arange(16, dtype=np.int32).reshape(1, 16)
        in1 = np.ones((1, 16), dtype=np.int32)
        in0_str = in0.astype(str).astype(np.object_)
        in1_str = in1.astype(str).astype(np.object_)
        inputs = [
            httpclient.InferInput("INPUT0", [1, 16], "BYTES"),
            httpclient.InferInput("INPUT1", [1, 16], "BYTES"),
        ]
        inputs[0].set_data_from_numpy(in0_str, binary_data=True)
        inputs[1].set_data_from_numpy(in1_str, binary_data=False)
        result = client.infer("simple_string", inputs)
        out0 = result.as_numpy("OUTPUT0")
        for i in range(16):
            assert int(out0[0][i]) == in0[0][i] + in1[0][i]
        print("PASS: string infer")
