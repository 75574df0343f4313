"""System shared-memory tests (CPU; model:
reference tests/test_shared_memory.py:34-183) + end-to-end shm
inference against the fixture server over HTTP and gRPC."""

import numpy as np
import pytest

import client_amd.http as httpclient
import client_amd.grpc as grpcclient
import client_amd.utils.shared_memory as shm
from client_amd.utils.shared_memory import SharedMemoryException


def test_create_set_get_destroy():
    handle = shm.create_shared_memory_region("t0", "/test_region_0", 64)
    try:
        data = np.arange(16, dtype=np.float32)
        shm.set_shared_memory_region(handle, [data])
        out = shm.get_contents_as_numpy(handle, np.float32, [16])
        np.testing.assert_array_equal(out, data)
        assert "/test_region_0" in shm.mapped_shared_memory_regions()
    finally:
        shm.destroy_shared_memory_region(handle)
    assert "/test_region_0" not in shm.mapped_shared_memory_regions()


def test_duplicate_key_refcount():
    h1 = shm.create_shared_memory_region("t1", "/test_dup", 64)
    h2 = shm.create_shared_memory_region("t1b", "/test_dup", 64)
    assert shm.mapped_shared_memory_regions().count("/test_dup") == 1
    shm.destroy_shared_memory_region(h1)
    # still mapped by h2
    assert "/test_dup" in shm.mapped_shared_memory_regions()
    shm.destroy_shared_memory_region(h2)
    assert "/test_dup" not in shm.mapped_shared_memory_regions()


def test_create_only_conflict():
    h1 = shm.create_shared_memory_region("t2", "/test_conflict", 64)
    try:
        with pytest.raises(SharedMemoryException):
            shm.create_shared_memory_region(
                "t2b", "/test_conflict", 64, create_only=True
            )
    finally:
        shm.destroy_shared_memory_region(h1)


def test_set_offset_and_bytes():
    handle = shm.create_shared_memory_region("t3", "/test_region_3", 256)
    try:
        s = np.array([b"hello", b"shm"], dtype=np.object_)
        shm.set_shared_memory_region(handle, [s], offset=8)
        out = shm.get_contents_as_numpy(handle, np.object_, [2], offset=8)
        np.testing.assert_array_equal(out, s)
    finally:
        shm.destroy_shared_memory_region(handle)


def test_destroy_unmapped_raises():
    handle = shm.create_shared_memory_region("t4", "/test_region_4", 64)
    shm.destroy_shared_memory_region(handle)
    with pytest.raises(SharedMemoryException):
        shm.destroy_shared_memory_region(handle)


@pytest.mark.parametrize("protocol", ["http", "grpc"])
def test_shm_inference_roundtrip(protocol, http_fixture_server,
                                 grpc_fixture_server):
    """Full data-plane round trip: inputs and outputs both in system shm;
    no tensor bytes on the wire (reference example: simple_http_shm_client)."""
    if protocol == "http":
        host, port, _ = http_fixture_server
        client = httpclient.InferenceServerClient(f"{host}:{port}")
        mod = httpclient
    else:
        host, port, _ = grpc_fixture_server
        client = grpcclient.InferenceServerClient(f"{host}:{port}")
        mod = grpcclient

    a = np.arange(16, dtype=np.int32).reshape(1, 16)
    b = np.full((1, 16), 5, dtype=np.int32)
    key = f"/shm_io_{protocol}"
    handle = shm.create_shared_memory_region("io", key, 256)
    try:
        shm.set_shared_memory_region(handle, [a, b])
        client.register_system_shared_memory("io", key, 256)
        status = client.get_system_shared_memory_status()
        if protocol == "http":
            assert any(r["name"] == "io" for r in status)
        else:
            assert "io" in status.regions

        inputs = [
            mod.InferInput("INPUT0", [1, 16], "INT32"),
            mod.InferInput("INPUT1", [1, 16], "INT32"),
        ]
        inputs[0].set_shared_memory("io", 64, offset=0)
        inputs[1].set_shared_memory("io", 64, offset=64)
        outputs = [
            mod.InferRequestedOutput("OUTPUT0"),
            mod.InferRequestedOutput("OUTPUT1"),
        ]
        outputs[0].set_shared_memory("io", 64, offset=128)
        outputs[1].set_shared_memory("io", 64, offset=192)
        result = client.infer("simple", inputs, outputs=outputs)
        # outputs are in the region, not on the wire
        assert result.as_numpy("OUTPUT0") is None or protocol == "http"
        out0 = shm.get_contents_as_numpy(handle, np.int32, [1, 16], offset=128)
        out1 = shm.get_contents_as_numpy(handle, np.int32, [1, 16], offset=192)
        np.testing.assert_array_equal(out0, a + b)
        np.testing.assert_array_equal(out1, a - b)
        client.unregister_system_shared_memory("io")
    finally:
        shm.destroy_shared_memory_region(handle)
        client.close()


def test_set_region_oversize_raises():
    """Writing past the region boundary is a SharedMemoryException, not a
    raw mmap IndexError (reference tests/test_shared_memory.py:92)."""
    import numpy as np

    import client_amd.utils.shared_memory as shm
    from client_amd.utils.shared_memory import SharedMemoryException

    handle = shm.create_shared_memory_region("over", "/shm_over_test", 32)
    try:
        big = np.zeros(64, dtype=np.int32)
        with pytest.raises(SharedMemoryException):
            shm.set_shared_memory_region(handle, [big])
        with pytest.raises(SharedMemoryException):
            shm.set_shared_memory_region(
                handle, [np.zeros(4, dtype=np.int32)], offset=24)
    finally:
        shm.destroy_shared_memory_region(handle)


def test_invalid_create_raises():
    """Zero/negative byte sizes are rejected up front (reference
    tests/test_shared_memory.py:65)."""
    import client_amd.utils.shared_memory as shm
    from client_amd.utils.shared_memory import SharedMemoryException

    with pytest.raises(SharedMemoryException):
        shm.create_shared_memory_region("bad", "/shm_bad_test", 0)
    with pytest.raises(SharedMemoryException):
        shm.create_shared_memory_region("bad", "/shm_bad_test", -4)
