#!/usr/bin/env python3
"""Stateful sequences over the bi-di stream — the server holds
per-sequence state (reference: simple_grpc_sequence_stream_infer_client.py)."""
import argparse
import queue

import numpy as np

import tritonclient.grpc as grpcclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()

    with grpcclient.InferenceServerClient(args.url) as client:
        results = queue.Queue()
        client.start_stream(callback=lambda result, error: results.put((result, error)))
        values = [11, 7, 5, 3, 2, 0, 1]
        for i, v in enumerate(values):
            inp = grpcclient.InferInput("INPUT", [1], "INT32")
            inp.set_data_from_numpy(np.array([v], dtype=np.int32))
            client.async_stream_infer(
                "sequence_accumulate", [inp], sequence_id=1007,
                sequence_start=(i == 0), sequence_end=(i == len(values) - 1))
        total = 0
        for v in values:
            total += v
            result, error = results.get(timeout=30)
            assert error is None
            assert int(result.as_numpy("OUTPUT")[0]) == total
        client.stop_stream()
        print("PASS: sequence stream")
