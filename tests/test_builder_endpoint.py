"""Fluent request builder (the Rust InferRequestBuilder surface,
reference src/rust/triton-client/src/infer.rs:548) and rotating
endpoints + retry (the Java endpoint package, reference
src/java/.../endpoint/AbstractEndpoint.java,
InferenceServerClient.java:245-374)."""

import numpy as np
import pytest

from client_amd import (
    InferRequestBuilder,
    MultiEndpointClient,
    RoundRobinEndpoint,
)
from client_amd import grpc as grpcclient
from client_amd import http as httpclient
from client_amd.utils import InferenceServerException


def test_builder_http_infer(http_fixture_server):
    host, port, _ = http_fixture_server
    client = httpclient.InferenceServerClient(f"{host}:{port}")
    try:
        x0 = np.arange(16, dtype=np.int32).reshape(1, 16)
        x1 = np.ones((1, 16), dtype=np.int32)
        result = (InferRequestBuilder("simple")
                  .request_id("b1")
                  .input_from_numpy("INPUT0", x0)
                  .input_from_numpy("INPUT1", x1)
                  .output("OUTPUT0")
                  .output("OUTPUT1")
                  .infer(client))
        np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x0 + x1)
        np.testing.assert_array_equal(result.as_numpy("OUTPUT1"), x0 - x1)
        assert result.get_response()["id"] == "b1"
    finally:
        client.close()


def test_builder_grpc_infer(grpc_fixture_server):
    host, port, _ = grpc_fixture_server
    client = grpcclient.InferenceServerClient(f"{host}:{port}")
    try:
        x0 = np.arange(16, dtype=np.int32).reshape(1, 16)
        x1 = np.full((1, 16), 3, dtype=np.int32)
        result = (InferRequestBuilder("simple")
                  .input_from_numpy("INPUT0", x0)
                  .input_from_numpy("INPUT1", x1)
                  .infer(client))
        np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x0 + x1)
    finally:
        client.close()


def test_builder_dtype_inference_and_bytes():
    b = (InferRequestBuilder("m")
         .input_from_numpy("A", np.zeros((2, 3), np.float32))
         .input_bytes("B", [b"x", "y"]))
    assert b._inputs[0][2][1] == "FP32"
    assert b._inputs[1][2][1] == "BYTES"


def test_round_robin_endpoint_rotation():
    ep = RoundRobinEndpoint(["a:1", "b:2", "c:3"])
    seen = [ep.get_next() for _ in range(6)]
    assert seen == ["a:1", "b:2", "c:3", "a:1", "b:2", "c:3"]
    assert ep.size() == 3


def test_multi_endpoint_rotates_and_fails_over(http_fixture_server):
    host, port, _ = http_fixture_server
    # one dead endpoint + the live fixture; retries must carry the
    # request to the live one
    dead = "127.0.0.1:1"  # port 1: connection refused
    client = MultiEndpointClient(
        [dead, f"{host}:{port}"], protocol="http", retries=2,
        network_timeout=5.0, connection_timeout=5.0,
    )
    try:
        # two calls: first rotates to the dead endpoint and retries to
        # the live one; second hits the live one directly
        assert client.is_server_live() is True
        assert client.is_server_live() is True
        x0 = np.arange(16, dtype=np.int32).reshape(1, 16)
        x1 = np.ones((1, 16), dtype=np.int32)
        result = (InferRequestBuilder("simple")
                  .input_from_numpy("INPUT0", x0)
                  .input_from_numpy("INPUT1", x1)
                  .infer(client))
        np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x0 + x1)
    finally:
        client.close()


def test_multi_endpoint_exhausted_raises():
    client = MultiEndpointClient(
        ["127.0.0.1:1", "127.0.0.1:2"], protocol="http", retries=1,
        network_timeout=2.0, connection_timeout=2.0,
    )
    try:
        with pytest.raises((InferenceServerException, OSError)):
            client.is_server_live()
    finally:
        client.close()


def test_multi_endpoint_grpc(grpc_fixture_server):
    host, port, _ = grpc_fixture_server
    client = MultiEndpointClient([f"{host}:{port}"], protocol="grpc")
    try:
        assert client.is_server_live() is True
        x0 = np.arange(16, dtype=np.int32).reshape(1, 16)
        x1 = np.full((1, 16), 2, dtype=np.int32)
        result = (InferRequestBuilder("simple")
                  .input_from_numpy("INPUT0", x0)
                  .input_from_numpy("INPUT1", x1)
                  .infer(client))
        np.testing.assert_array_equal(result.as_numpy("OUTPUT1"), x0 - x1)
    finally:
        client.close()


def test_builder_resolves_aio_client_classes():
    """The builder's protocol resolution also covers the aio clients
    (module path contains .http/.grpc)."""
    import asyncio

    from client_amd._builder import _io_classes_for
    from client_amd.http import InferInput as HttpInput
    from client_amd.http.aio import InferenceServerClient as AioHttp

    async def make():
        return AioHttp("127.0.0.1:1")

    client = asyncio.new_event_loop().run_until_complete(make())
    try:
        input_cls, _ = _io_classes_for(client)
        assert input_cls is HttpInput
    finally:
        pass


def test_builder_shared_memory_bindings():
    b = (InferRequestBuilder("m")
         .shared_memory_input("IN", "region_a", 1024, [4, 64], "FP32",
                              offset=128)
         .shared_memory_output("OUT", "region_b", 2048))
    from client_amd import http as httpclient

    class FakeHttpClient(httpclient.InferenceServerClient):
        protocol = "http"  # builder resolution hint (module is the
        #                    test file, not client_amd.http)

        def __init__(self):  # no network
            pass

    inputs, outputs, kwargs = b.build(FakeHttpClient())
    t = inputs[0]._get_tensor()
    assert t["parameters"]["shared_memory_region"] == "region_a"
    assert t["parameters"]["shared_memory_offset"] == 128
    assert t["shape"] == [4, 64]
