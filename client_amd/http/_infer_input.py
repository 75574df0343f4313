"""HTTP InferInput (reference: tritonclient/http/_infer_input.py).

Holds one named input tensor for a KServe-v2 HTTP inference request:
either inline JSON data, raw binary data, or a shared-memory reference.
"""

import numpy as np

from ..utils import (
    np_to_triton_dtype,
    raise_error,
    serialize_bf16_tensor,
    serialize_byte_tensor,
)


class InferInput:
    def __init__(self, name, shape, datatype):
        self._name = name
        self._shape = list(shape)
        self._datatype = datatype
        self._parameters = {}
        self._data = None
        self._raw_data = None

    def name(self):
        return self._name

    def datatype(self):
        return self._datatype

    def shape(self):
        return self._shape

    def set_shape(self, shape):
        self._shape = list(shape)
        return self

    def _validate_array(self, input_tensor):
        """Dtype/shape admission for set_data_from_numpy. The error
        strings are part of the compat contract (user code matches on
        them); tests/test_infer_input_compat.py pins them against the
        reference wording (reference http/_infer_input.py:130-160)."""
        if not isinstance(input_tensor, np.ndarray):
            raise_error("input_tensor must be a numpy array")
        got = np_to_triton_dtype(input_tensor.dtype)
        if got != self._datatype:
            # a BF16 input has no numpy dtype; it is fed as fp32 (or
            # fp16) and truncated on serialization
            if self._datatype == "BF16" and input_tensor.dtype in (
                np.float16, np.float32,
            ):
                pass
            elif self._datatype == "BF16":
                raise_error(
                    "got unexpected datatype {} from numpy array, expected "
                    "{} for BF16 input".format(got, np.float32)
                )
            else:
                raise_error(
                    "got unexpected datatype {} from numpy array, "
                    "expected {}".format(got, self._datatype)
                )
        if tuple(input_tensor.shape) != tuple(self._shape):
            raise_error(
                "got unexpected numpy array shape [{}], expected [{}]".format(
                    str(input_tensor.shape)[1:-1], str(self._shape)[1:-1]
                )
            )

    @staticmethod
    def _bytes_elements_as_json(input_tensor):
        """Flatten a BYTES tensor to a list of JSON-safe strings:
        bytes elements are UTF-8 decoded, anything else is str()'d."""
        if input_tensor.size == 0:
            return []
        out = []
        for elem in input_tensor.reshape(-1, order="C").tolist():
            if isinstance(elem, bytes):
                try:
                    out.append(elem.decode("utf-8"))
                except UnicodeDecodeError:
                    raise_error(
                        'Failed to encode "{}" using UTF-8. Please use '
                        "binary_data=True, if you want to pass a byte "
                        "array.".format(elem)
                    )
            else:
                out.append(str(elem))
        return out

    def set_data_from_numpy(self, input_tensor, binary_data=True):
        """Attach tensor data from a numpy array.

        binary_data=True sends the raw little-endian bytes after the
        JSON header; False inlines the values into the JSON ``data``
        field. Only BF16 is binary-only (it has no JSON representation;
        FP16 inlines fine — reference _infer_input.py:169-171 restricts
        exactly BF16).
        """
        self._validate_array(input_tensor)

        # numpy data replaces any shared-memory binding
        for key in ("shared_memory_region", "shared_memory_byte_size",
                    "shared_memory_offset"):
            self._parameters.pop(key, None)

        if binary_data:
            self._data = None
            if self._datatype == "BYTES":
                packed = serialize_byte_tensor(input_tensor)
                self._raw_data = packed.item() if packed.size else b""
            elif self._datatype == "BF16":
                packed = serialize_bf16_tensor(input_tensor)
                self._raw_data = packed.item() if packed.size else b""
            else:
                self._raw_data = input_tensor.tobytes()
            self._parameters["binary_data_size"] = len(self._raw_data)
        else:
            if self._datatype == "BF16":
                raise_error(
                    "BF16 inputs must be sent as binary data over HTTP. "
                    "Please set binary_data=True"
                )
            self._raw_data = None
            self._parameters.pop("binary_data_size", None)
            if self._datatype == "BYTES":
                self._data = self._bytes_elements_as_json(input_tensor)
            else:
                self._data = input_tensor.reshape(-1, order="C").tolist()
        return self

    def set_shared_memory(self, region_name, byte_size, offset=0):
        """Reference a (system or HIP) shared-memory region instead of
        sending tensor bytes on the wire."""
        self._data = None
        self._raw_data = None
        self._parameters.pop("binary_data_size", None)
        self._parameters["shared_memory_region"] = region_name
        self._parameters["shared_memory_byte_size"] = byte_size
        if offset != 0:
            self._parameters["shared_memory_offset"] = offset
        return self

    def _get_binary_data(self):
        return self._raw_data

    def _get_tensor(self):
        """The JSON dict for this input in the request header."""
        tensor = {
            "name": self._name,
            "shape": self._shape,
            "datatype": self._datatype,
        }
        if self._parameters:
            tensor["parameters"] = self._parameters
        if self._parameters.get("shared_memory_region") is None and self._raw_data is None:
            if self._data is not None:
                tensor["data"] = self._data
        return tensor
