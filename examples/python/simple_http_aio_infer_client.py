#!/usr/bin/env python3
"""asyncio HTTP inference (reference: simple_http_aio_infer_client.py)."""
import argparse
import asyncio

import numpy as np

import tritonclient.http.aio as aioclient


async def main(url):
    async with aioclient.InferenceServerClient(url) as client:
        assert await client.is_server_live()
        inputs = [
            aioclient.InferInput("INPUT0", [1, 16], "INT32"),
            aioclient.InferInput("INPUT1", [1, 16], "INT32"),
        ]
        in0 = np.arange(16, dtype=np.int32).reshape(1, 16)
        in1 = np.ones((1, 16), dtype=np.int32)
        inputs[0].set_data_from_numpy(in0)
        inputs[1].set_data_from_numpy(in1)
        result = await client.infer("simple", inputs)
        assert (result.as_numpy("OUTPUT0") == in0 + in1).all()
        print("PASS: aio infer")


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8000")
    args = parser.parse_args()
    asyncio.run(main(args.url))
