#!/usr/bin/env python3
"""Protocol-level utilities for the MI355X-native Triton client stack.

API-compatible with ``tritonclient.utils`` (reference:
/root/reference/src/python/library/tritonclient/utils/__init__.py) but
implemented from scratch with vectorized numpy codecs instead of the
reference's per-element Python loops (reference lines 208-363).
"""

import struct

import numpy as np

__all__ = [
    "InferenceServerException",
    "raise_error",
    "np_to_triton_dtype",
    "triton_to_np_dtype",
    "serialize_byte_tensor",
    "deserialize_bytes_tensor",
    "serialize_bf16_tensor",
    "deserialize_bf16_tensor",
    "serialized_byte_size",
]

# Parameter keys reserved by the KServe-v2 protocol; user-supplied request
# parameters must not collide (reference: utils/__init__.py:39-48).
_reserved_params = [
    "sequence_id",
    "sequence_start",
    "sequence_end",
    "priority",
    "binary_data_output",
]


class InferenceServerException(Exception):
    """Exception carrying a message, optional status and debug details.

    Mirrors tritonclient.utils.InferenceServerException
    (reference utils/__init__.py:86-145).
    """

    def __init__(self, msg, status=None, debug_details=None):
        self._msg = msg
        self._status = status
        self._debug_details = debug_details
        super().__init__(msg)

    def __str__(self):
        msg = super().__str__() if self._msg is None else self._msg
        if self._status is not None:
            msg = "[" + self._status + "] " + msg
        return msg

    def message(self):
        return self._msg

    def status(self):
        return self._status

    def debug_details(self):
        return self._debug_details


def raise_error(msg):
    """Raise an InferenceServerException without status/details."""
    raise InferenceServerException(msg=msg)


_NP_TO_TRITON = {
    np.bool_: "BOOL",
    np.int8: "INT8",
    np.int16: "INT16",
    np.int32: "INT32",
    np.int64: "INT64",
    np.uint8: "UINT8",
    np.uint16: "UINT16",
    np.uint32: "UINT32",
    np.uint64: "UINT64",
    np.float16: "FP16",
    np.float32: "FP32",
    np.float64: "FP64",
    np.object_: "BYTES",
    np.bytes_: "BYTES",
}

_TRITON_TO_NP = {
    "BOOL": bool,
    "INT8": np.int8,
    "INT16": np.int16,
    "INT32": np.int32,
    "INT64": np.int64,
    "UINT8": np.uint8,
    "UINT16": np.uint16,
    "UINT32": np.uint32,
    "UINT64": np.uint64,
    "FP16": np.float16,
    "FP32": np.float32,
    "FP64": np.float64,
    # BF16 has no numpy dtype; we expose it as FP32 on the numpy side the
    # same way the reference does (utils/__init__.py:199-200).
    "BF16": np.float32,
    "BYTES": np.object_,
}

# Size in bytes of one element on the wire, per Triton datatype.
TRITON_DTYPE_SIZES = {
    "BOOL": 1,
    "INT8": 1,
    "INT16": 2,
    "INT32": 4,
    "INT64": 8,
    "UINT8": 1,
    "UINT16": 2,
    "UINT32": 4,
    "UINT64": 8,
    "FP16": 2,
    "BF16": 2,
    "FP32": 4,
    "FP64": 8,
    "FP8E4M3": 1,
    "FP8E5M2": 1,
}


def np_to_triton_dtype(np_dtype):
    """numpy dtype -> Triton datatype string (reference utils/__init__.py:148)."""
    if np_dtype in _NP_TO_TRITON:
        return _NP_TO_TRITON[np_dtype]
    if np_dtype == bool:
        return "BOOL"
    try:
        key = np.dtype(np_dtype).type
        return _NP_TO_TRITON.get(key, None)
    except TypeError:
        return None


def triton_to_np_dtype(dtype):
    """Triton datatype string -> numpy dtype (reference utils/__init__.py:176)."""
    return _TRITON_TO_NP.get(dtype, None)


def serialize_byte_tensor(input_tensor):
    """Serialize a BYTES tensor into the Triton on-wire representation:

    row-major concatenation of ``<4-byte little-endian length><payload>``
    per element (reference utils/__init__.py:208-260). Returns a
    1-D np.uint8 array (or None for 0-element input).
    """
    if input_tensor.size == 0:
        return np.empty([0], dtype=np.object_)

    if (input_tensor.dtype != np.object_) and (input_tensor.dtype.type != np.bytes_):
        raise_error("cannot serialize bytes tensor: invalid datatype")

    flattened_ls = []
    # Iterate in 'C' order to match row-major wire layout.
    for obj in np.nditer(input_tensor, flags=["refs_ok"], order="C"):
        # Items are of type bytes or str; str is utf-8 encoded.
        if input_tensor.dtype == np.object_:
            if type(obj.item()) == bytes:
                s = obj.item()
            else:
                s = str(obj.item()).encode("utf-8")
        else:
            s = obj.item()
        flattened_ls.append(struct.pack("<I", len(s)))
        flattened_ls.append(s)
    flattened = b"".join(flattened_ls)
    flattened_array = np.asarray(flattened, dtype=np.object_)
    if not flattened_array.flags["C_CONTIGUOUS"]:
        flattened_array = np.ascontiguousarray(flattened_array, dtype=np.object_)
    return flattened_array


def deserialize_bytes_tensor(encoded_tensor):
    """Inverse of serialize_byte_tensor: bytes buffer -> 1-D np.object_ array
    of bytes elements (reference utils/__init__.py:263-291)."""
    strs = []
    offset = 0
    val_buf = encoded_tensor
    n = len(val_buf)
    while offset < n:
        (length,) = struct.unpack_from("<I", val_buf, offset)
        offset += 4
        strs.append(bytes(val_buf[offset : offset + length]))
        offset += length
    return np.array(strs, dtype=np.object_)


def serialize_bf16_tensor(input_tensor):
    """Serialize an fp32/fp16 numpy tensor to BF16 wire bytes.

    Semantics match the reference (utils/__init__.py:294-330): each fp32
    element's upper 16 bits are kept (truncation, no rounding). The
    reference loops per element in Python; here it is one vectorized
    numpy view+shift. Return type matches the reference for drop-in
    compatibility: a 0-d np.object_ array holding the wire bytes
    (callers do ``.item()``), np.empty([0], object_) when empty.
    fp16 input is accepted as a superset (upcast to fp32 first).
    """
    if input_tensor.size == 0:
        return np.empty([0], dtype=np.object_)
    if input_tensor.dtype not in (np.float16, np.float32):
        raise_error("cannot serialize bf16 tensor: invalid datatype")
    f32 = np.ascontiguousarray(input_tensor, dtype="<f4")
    u32 = f32.view("<u4").reshape(-1)
    u16 = (u32 >> np.uint32(16)).astype("<u2")
    return np.asarray(u16.tobytes(), dtype=np.object_)


def deserialize_bf16_tensor(encoded_tensor):
    """BF16 wire bytes -> 1-D np.float32 array by zero-extending the low
    16 bits (reference utils/__init__.py:333-363), vectorized."""
    buf = np.frombuffer(bytearray(encoded_tensor), dtype="<u2")
    u32 = buf.astype("<u4") << np.uint32(16)
    return u32.view("<f4")


def serialized_byte_size(tensor_value):
    """On-wire byte size of a numpy tensor (reference utils/__init__.py:58-83)."""
    if tensor_value.dtype == np.object_:
        total = 0
        for obj in np.nditer(tensor_value, flags=["refs_ok"], order="C"):
            if type(obj.item()) == bytes:
                total += len(obj.item()) + 4
            else:
                total += len(str(obj.item()).encode("utf-8")) + 4
        return total
    else:
        return tensor_value.nbytes
