"""Unit tests for client_amd.utils codecs (model:
reference tests/test_inference_server_client.py + utils semantics)."""

import numpy as np
import pytest

from client_amd.utils import (
    InferenceServerException,
    deserialize_bf16_tensor,
    deserialize_bytes_tensor,
    np_to_triton_dtype,
    serialize_bf16_tensor,
    serialize_byte_tensor,
    serialized_byte_size,
    triton_to_np_dtype,
)


def test_dtype_roundtrip():
    pairs = [
        (np.bool_, "BOOL"), (np.int8, "INT8"), (np.int16, "INT16"),
        (np.int32, "INT32"), (np.int64, "INT64"), (np.uint8, "UINT8"),
        (np.uint16, "UINT16"), (np.uint32, "UINT32"), (np.uint64, "UINT64"),
        (np.float16, "FP16"), (np.float32, "FP32"), (np.float64, "FP64"),
        (np.object_, "BYTES"),
    ]
    for np_dt, tr in pairs:
        assert np_to_triton_dtype(np_dt) == tr
    assert triton_to_np_dtype("FP32") == np.float32
    assert triton_to_np_dtype("BF16") == np.float32
    assert triton_to_np_dtype("BYTES") == np.object_
    assert np_to_triton_dtype(np.dtype("float32")) == "FP32"


def test_bytes_tensor_roundtrip():
    arr = np.array([b"hello", b"", b"world \xff\x00binary", "unicode é".encode()],
                   dtype=np.object_).reshape(2, 2)
    serialized = serialize_byte_tensor(arr)
    raw = serialized.item()
    # wire layout: 4-byte LE length + payload, row-major
    assert raw[:4] == (5).to_bytes(4, "little")
    out = deserialize_bytes_tensor(raw)
    assert out.shape == (4,)
    assert list(out) == [b"hello", b"", b"world \xff\x00binary", "unicode é".encode()]


def test_bytes_tensor_strings():
    arr = np.array(["abc", "defg"], dtype=np.object_)
    raw = serialize_byte_tensor(arr).item()
    out = deserialize_bytes_tensor(raw)
    assert list(out) == [b"abc", b"defg"]


def test_serialized_byte_size():
    arr = np.array([b"abc", b"de"], dtype=np.object_)
    assert serialized_byte_size(arr) == 3 + 4 + 2 + 4
    arr2 = np.zeros((3, 4), dtype=np.float32)
    assert serialized_byte_size(arr2) == 48


def test_bf16_roundtrip_exact():
    # Values exactly representable in bf16 survive the round trip.
    vals = np.array([1.0, -2.5, 0.0, 0.5, -0.375, 128.0], dtype=np.float32)
    raw = serialize_bf16_tensor(vals)
    # reference-compatible return type: 0-d object_ array, .item() gives
    # the wire bytes (reference utils/__init__.py:294-330 callers)
    assert raw.dtype == np.object_
    wire = raw.item()
    assert isinstance(wire, bytes) and len(wire) == vals.size * 2
    back = deserialize_bf16_tensor(wire)
    np.testing.assert_array_equal(back, vals)


def test_bf16_empty_returns_empty_object_array():
    raw = serialize_bf16_tensor(np.empty((0,), dtype=np.float32))
    assert raw.dtype == np.object_ and raw.size == 0


def test_bf16_truncation_semantics():
    # Reference semantics: plain truncation of the fp32 low 16 bits
    # (no round-to-nearest-even) — utils/__init__.py:294-330.
    x = np.array([1.0000001], dtype=np.float32)
    wire = serialize_bf16_tensor(x).item()
    expected = x.view(np.uint32) >> 16
    assert np.frombuffer(wire, dtype=np.uint16)[0] == expected[0]
    back = deserialize_bf16_tensor(wire)
    # truncation error is bounded by 1 ulp of bf16
    assert abs(back[0] - x[0]) < 2 ** -7


def test_bf16_from_fp16():
    x = np.array([1.5, -3.25], dtype=np.float16)
    wire = serialize_bf16_tensor(x).item()
    back = deserialize_bf16_tensor(wire)
    np.testing.assert_allclose(back, x.astype(np.float32), rtol=2 ** -7)


def test_bf16_invalid_dtype():
    with pytest.raises(InferenceServerException):
        serialize_bf16_tensor(np.zeros(3, dtype=np.int32))


def test_exception_fields():
    e = InferenceServerException("msg", "400", "dbg")
    assert e.message() == "msg"
    assert e.status() == "400"
    assert e.debug_details() == "dbg"
    assert "[400] msg" == str(e)
