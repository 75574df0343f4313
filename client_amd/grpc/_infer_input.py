"""gRPC InferInput (reference: tritonclient/grpc/_infer_input.py).

Tensor bytes ride in ModelInferRequest.raw_input_contents — the wire
form the protocol chose explicitly for performance
(grpc_service.proto:683-706).
"""

import numpy as np

from ..utils import (
    np_to_triton_dtype,
    raise_error,
    serialize_bf16_tensor,
    serialize_byte_tensor,
)
from ._proto import service_pb2


class InferInput:
    def __init__(self, name, shape, datatype):
        self._input = service_pb2.ModelInferRequest.InferInputTensor()
        self._input.name = name
        self._input.ClearField("shape")
        self._input.shape.extend(shape)
        self._input.datatype = datatype
        self._raw_content = None

    def name(self):
        return self._input.name

    def datatype(self):
        return self._input.datatype

    def shape(self):
        return list(self._input.shape)

    def set_shape(self, shape):
        self._input.ClearField("shape")
        self._input.shape.extend(shape)
        return self

    def _validate_array(self, input_tensor):
        """Dtype/shape admission; error strings are compat contract,
        pinned in tests/test_infer_input_compat.py (reference
        grpc/_infer_input.py wording)."""
        if not isinstance(input_tensor, np.ndarray):
            raise_error("input_tensor must be a numpy array")
        got = np_to_triton_dtype(input_tensor.dtype)
        if got != self._input.datatype:
            # BF16 has no numpy dtype: fp32 (or fp16) in, truncated on
            # serialization
            if self._input.datatype == "BF16" and input_tensor.dtype in (
                np.float16, np.float32,
            ):
                pass
            elif self._input.datatype == "BF16":
                raise_error(
                    "got unexpected datatype {} from numpy array, expected "
                    "float16/float32 for BF16 input".format(got)
                )
            else:
                raise_error(
                    "got unexpected datatype {} from numpy array, "
                    "expected {}".format(got, self._input.datatype)
                )
        if tuple(input_tensor.shape) != tuple(self._input.shape):
            raise_error(
                "got unexpected numpy array shape [{}], expected [{}]".format(
                    str(input_tensor.shape)[1:-1],
                    str(list(self._input.shape))[1:-1],
                )
            )

    def set_data_from_numpy(self, input_tensor):
        """Serialize a numpy array into this input's raw wire bytes
        (rides in ModelInferRequest.raw_input_contents)."""
        self._validate_array(input_tensor)
        # numpy data replaces any shared-memory binding
        for key in ("shared_memory_region", "shared_memory_byte_size",
                    "shared_memory_offset"):
            self._input.parameters.pop(key, None)
        dt = self._input.datatype
        if dt in ("BYTES", "BF16"):
            serializer = (serialize_byte_tensor if dt == "BYTES"
                          else serialize_bf16_tensor)
            packed = serializer(input_tensor)
            self._raw_content = packed.item() if packed.size else b""
        else:
            self._raw_content = np.ascontiguousarray(input_tensor).tobytes()
        return self

    def set_raw_bytes(self, raw_bytes):
        """Attach pre-serialized tensor bytes directly (zero extra copies);
        used by the device pack path and the load generator."""
        self._raw_content = raw_bytes
        return self

    def set_shared_memory(self, region_name, byte_size, offset=0):
        self._input.ClearField("contents")
        self._raw_content = None
        self._input.parameters["shared_memory_region"].string_param = region_name
        self._input.parameters["shared_memory_byte_size"].int64_param = byte_size
        if offset != 0:
            self._input.parameters["shared_memory_offset"].int64_param = offset
        return self

    def _get_tensor(self):
        return self._input

    def _get_content(self):
        return self._raw_content
