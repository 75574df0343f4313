#!/usr/bin/env python3
"""Stateful sequences with SYNC HTTP infer calls
(reference: simple_http_sequence_sync_infer_client.py)."""
import argparse

import numpy as np

import tritonclient.http as httpclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8000")
    args = parser.parse_args()

    with httpclient.InferenceServerClient(args.url) as client:
        values = [4, 2, 0, 1]
        total = 0
        for i, v in enumerate(values):
            inp = httpclient.InferInput("INPUT", [1], "INT32")
            inp.set_data_from_numpy(np.array([v], dtype=np.int32))
            result = client.infer(
                "sequence_accumulate", [inp], sequence_id=43,
                sequence_start=(i == 0),
                sequence_end=(i == len(values) - 1),
            )
            total += v
            assert int(result.as_numpy("OUTPUT")[0]) == total
        print("PASS: http sequence sync")
