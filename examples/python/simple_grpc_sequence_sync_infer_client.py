#!/usr/bin/env python3
"""Stateful sequences with SYNC infer calls — sequence_id/start/end
ride as request parameters, no stream needed
(reference: simple_grpc_sequence_sync_infer_client.py)."""
import argparse

import numpy as np

import tritonclient.grpc as grpcclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()

    with grpcclient.InferenceServerClient(args.url) as client:
        values = [4, 2, 0, 1]
        total = 0
        for i, v in enumerate(values):
            inp = grpcclient.InferInput("INPUT", [1], "INT32")
            inp.set_data_from_numpy(np.array([v], dtype=np.int32))
            result = client.infer(
                "sequence_accumulate", [inp], sequence_id=42,
                sequence_start=(i == 0),
                sequence_end=(i == len(values) - 1),
            )
            total += v
            assert int(result.as_numpy("OUTPUT")[0]) == total
        print("PASS: grpc sequence sync")
