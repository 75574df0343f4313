#!/usr/bin/env python3
"""Stateful sequences over the asyncio bi-di stream
(reference: simple_grpc_aio_sequence_stream_infer_client.py)."""
import argparse
import asyncio

import numpy as np

import tritonclient.grpc.aio as grpcclient


async def main(url):
    async with grpcclient.InferenceServerClient(url) as client:
        values = [11, 7, 5, 3]

        async def requests():
            for i, v in enumerate(values):
                inp = grpcclient.InferInput("INPUT", [1], "INT32")
                inp.set_data_from_numpy(np.array([v], dtype=np.int32))
                yield {
                    "model_name": "sequence_accumulate",
                    "inputs": [inp],
                    "sequence_id": 1009,
                    "sequence_start": i == 0,
                    "sequence_end": i == len(values) - 1,
                }

        total = 0
        i = 0
        async for result, error in client.stream_infer(requests()):
            assert error is None, error
            total += values[i]
            assert int(result.as_numpy("OUTPUT")[0]) == total
            i += 1
        assert i == len(values)
        print("PASS: aio sequence stream")


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()
    asyncio.run(main(args.url))
