#!/usr/bin/env python3
"""Reuse InferInput/InferRequestedOutput objects across requests
(reference: reuse_infer_objects_client.py)."""
import argparse

import numpy as np

import tritonclient.http as httpclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8000")
    args = parser.parse_args()

    with httpclient.InferenceServerClient(args.url) as client:
        inputs = [
            httpclient.InferInput("INPUT0", [1, 16], "INT32"),
            httpclient.InferInput("INPUT1", [1, 16], "INT32"),
        ]
        outputs = [httpclient.InferRequestedOutput("OUTPUT0")]
        for trial in range(3):
            in0 = np.full((1, 16), trial, dtype=np.int32)
            in1 = np.ones((1, 16), dtype=np.int32)
            inputs[0].set_data_from_numpy(in0)
            inputs[1].set_data_from_numpy(in1)
            result = client.infer("simple", inputs, outputs=outputs)
            assert (result.as_numpy("OUTPUT0") == in0 + in1).all()
        print("PASS: reuse infer objects")
