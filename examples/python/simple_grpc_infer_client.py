#!/usr/bin/env python3
"""Sync gRPC inference (reference: simple_grpc_infer_client.py)."""
import argparse

import numpy as np

import tritonclient.grpc as grpcclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()

    with grpcclient.InferenceServerClient(args.url) as client:
        inputs = [
            grpcclient.InferInput("INPUT0", [1, 16], "INT32"),
            grpcclient.InferInput("INPUT1", [1, 16], "INT32"),
        ]
        in0 = np.arange(16, dtype=np.int32).reshape(1, 16)
        in1 = np.ones((1, 16), dtype=np.int32)
        inputs[0].set_data_from_numpy(in0)
        inputs[1].set_data_from_numpy(in1)
        outputs = [
            grpcclient.InferRequestedOutput("OUTPUT0"),
            grpcclient.InferRequestedOutput("OUTPUT1"),
        ]
        result = client.infer("simple", inputs, outputs=outputs)
        assert (result.as_numpy("OUTPUT0") == in0 + in1).all()
        assert (result.as_numpy("OUTPUT1") == in0 - in1).all()
        print("PASS: grpc infer")
