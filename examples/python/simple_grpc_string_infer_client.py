#!/usr/bin/env python3
"""BYTES/string tensors over gRPC (reference: simple_grpc_string_infer_client.py)."""
import argparse

import numpy as np

import tritonclient.grpc as grpcclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()

    with grpcclient.InferenceServerClient(args.url) as client:
        in0 = np.array([str(i).encode() for i in range(16)],
                       dtype=np.object_).reshape(1, 16)
        in1 = np.array([b"1"] * 16, dtype=np.object_).reshape(1, 16)
        inputs = [
            grpcclient.InferInput("INPUT0", [1, 16], "BYTES"),
            grpcclient.InferInput("INPUT1", [1, 16], "BYTES"),
        ]
        inputs[0].set_data_from_numpy(in0)
        inputs[1].set_data_from_numpy(in1)
        result = client.infer("simple_string", inputs)
        out0 = result.as_numpy("OUTPUT0")
        expect = np.array([int(a) + 1 for a in range(16)])
        got = np.array([int(v) for v in out0.reshape(-1)])
        assert (got == expect).all(), (got, expect)
        print("PASS: grpc string infer")
