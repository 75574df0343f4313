"""asyncio client tests (http.aio + grpc.aio), CPU only."""

import asyncio

import numpy as np
import pytest


def _run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


def test_http_aio(http_fixture_server):
    import client_amd.http.aio as aioclient

    host, port, _ = http_fixture_server

    async def main():
        async with aioclient.InferenceServerClient(f"{host}:{port}") as client:
            assert await client.is_server_live()
            assert await client.is_server_ready()
            assert await client.is_model_ready("simple")
            meta = await client.get_server_metadata()
            assert meta["name"] == "client_amd_server"
            a = np.arange(16, dtype=np.int32).reshape(1, 16)
            b = np.ones((1, 16), dtype=np.int32)
            inputs = [
                aioclient.InferInput("INPUT0", [1, 16], "INT32"),
                aioclient.InferInput("INPUT1", [1, 16], "INT32"),
            ]
            inputs[0].set_data_from_numpy(a)
            inputs[1].set_data_from_numpy(b)
            result = await client.infer("simple", inputs)
            np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), a + b)
            stats = await client.get_inference_statistics("simple")
            assert stats["model_stats"][0]["inference_count"] >= 1

    _run(main())


def test_grpc_aio(grpc_fixture_server):
    import client_amd.grpc.aio as aioclient

    host, port, _ = grpc_fixture_server

    async def main():
        async with aioclient.InferenceServerClient(f"{host}:{port}") as client:
            assert await client.is_server_live()
            assert await client.is_model_ready("simple")
            meta = await client.get_server_metadata(as_json=True)
            assert meta["name"] == "client_amd_server"
            a = np.arange(16, dtype=np.int32).reshape(1, 16)
            inputs = [
                aioclient.InferInput("INPUT0", [1, 16], "INT32"),
                aioclient.InferInput("INPUT1", [1, 16], "INT32"),
            ]
            inputs[0].set_data_from_numpy(a)
            inputs[1].set_data_from_numpy(a)
            result = await client.infer("simple", inputs)
            np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), a + a)

    _run(main())


def test_grpc_aio_stream(grpc_fixture_server):
    import client_amd.grpc.aio as aioclient

    host, port, _ = grpc_fixture_server

    async def main():
        async with aioclient.InferenceServerClient(f"{host}:{port}") as client:
            async def requests():
                for i, (start, end) in enumerate(
                    [(True, False), (False, False), (False, True)]
                ):
                    inp = aioclient.InferInput("INPUT", [1], "INT32")
                    inp.set_data_from_numpy(np.array([i + 1], dtype=np.int32))
                    yield {
                        "model_name": "sequence_accumulate",
                        "inputs": [inp],
                        "sequence_id": 11,
                        "sequence_start": start,
                        "sequence_end": end,
                    }

            vals = []
            it = client.stream_infer(requests())
            async for result, error in it:
                assert error is None
                vals.append(int(result.as_numpy("OUTPUT")[0]))
                if len(vals) == 3:
                    break
            assert vals == [1, 3, 6]

    _run(main())


def test_http_aio_response_compression(http_fixture_server):
    """aio response_compression_algorithm: opt-in Accept-Encoding with
    manual whole-body decompression in InferResult."""
    import asyncio

    import numpy as np

    import client_amd.http.aio as aiohttpclient

    host, port, _ = http_fixture_server

    async def run():
        client = aiohttpclient.InferenceServerClient(f"{host}:{port}")
        try:
            x = np.random.rand(1, 1024).astype(np.float32)
            inp = aiohttpclient.InferInput("INPUT0", list(x.shape), "FP32")
            inp.set_data_from_numpy(x)
            for algo in ("gzip", "deflate"):
                result = await client.infer(
                    "identity_fp32", [inp],
                    response_compression_algorithm=algo,
                )
                np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x)
        finally:
            await client.close()

    asyncio.run(run())


def test_grpc_aio_stream_cancel(grpc_fixture_server):
    """aio stream iterator .cancel() aborts a slow decoupled request."""
    import asyncio
    import time

    import numpy as np

    import client_amd.grpc.aio as aiogrpc

    host, port, _ = grpc_fixture_server

    async def run():
        client = aiogrpc.InferenceServerClient(f"{host}:{port}")
        try:
            async def requests():
                inputs = [
                    aiogrpc.InferInput("IN", [4], "INT32"),
                    aiogrpc.InferInput("DELAY", [4], "UINT32"),
                ]
                inputs[0].set_data_from_numpy(np.arange(4, dtype=np.int32))
                inputs[1].set_data_from_numpy(
                    np.full(4, 800, dtype=np.uint32))
                yield {"model_name": "repeat_int32", "inputs": inputs}

            it = client.stream_infer(requests())
            got_one = False
            t0 = time.monotonic()
            async for result, error in it:
                assert error is None
                got_one = True
                it.cancel()
                break
            assert got_one
            assert time.monotonic() - t0 < 5.0
        finally:
            await client.close()

    asyncio.run(run())
