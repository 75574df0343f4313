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

    def set_data_from_numpy(self, input_tensor, binary_data=True):
        """Attach tensor data from a numpy array.

        binary_data=True sends the raw little-endian bytes after the JSON
        header; False inlines the values into the JSON ``data`` field
        (BF16 must be binary — reference _infer_input.py:106-242).
        """
        if not isinstance(input_tensor, (np.ndarray,)):
            raise_error("input_tensor must be a numpy array")

        dtype = np_to_triton_dtype(input_tensor.dtype)
        if self._datatype != dtype:
            if self._datatype == "BF16":
                if input_tensor.dtype not in (np.float16, np.float32):
                    raise_error(
                        "got unexpected datatype {} from numpy array, expected "
                        "{} for BF16 input".format(dtype, np.float32)
                    )
            else:
                raise_error(
                    "got unexpected datatype {} from numpy array, expected {}".format(
                        dtype, self._datatype
                    )
                )
        valid_shape = True
        if len(self._shape) != len(input_tensor.shape):
            valid_shape = False
        else:
            for i in range(len(self._shape)):
                if self._shape[i] != input_tensor.shape[i]:
                    valid_shape = False
        if not valid_shape:
            raise_error(
                "got unexpected numpy array shape [{}], expected [{}]".format(
                    str(input_tensor.shape)[1:-1], str(self._shape)[1:-1]
                )
            )

        self._parameters.pop("shared_memory_region", None)
        self._parameters.pop("shared_memory_byte_size", None)
        self._parameters.pop("shared_memory_offset", None)

        if not binary_data:
            if self._datatype in ("BF16", "FP16"):
                raise_error(
                    f"{self._datatype} inputs must be sent as binary data "
                    "over HTTP. Please "
                    "set binary_data=True"
                )
            self._parameters.pop("binary_data_size", None)
            self._raw_data = None
            if self._datatype == "BYTES":
                self._data = []
                try:
                    if input_tensor.size > 0:
                        for obj in np.nditer(
                            input_tensor, flags=["refs_ok"], order="C"
                        ):
                            # We need to convert the object to string using
                            # utf-8 codec.
                            if input_tensor.dtype == np.object_:
                                if type(obj.item()) == bytes:
                                    self._data.append(str(obj.item(), encoding="utf-8"))
                                else:
                                    self._data.append(str(obj.item()))
                            else:
                                self._data.append(str(obj.item(), encoding="utf-8"))
                except UnicodeDecodeError:
                    raise_error(
                        'Failed to encode "{}" using UTF-8. Please use '
                        "binary_data=True, if you want to pass a byte array.".format(
                            obj.item()
                        )
                    )
            else:
                self._data = [val.item() for val in input_tensor.flatten()]
        else:
            self._data = None
            if self._datatype == "BYTES":
                serialized_output = serialize_byte_tensor(input_tensor)
                if serialized_output.size > 0:
                    self._raw_data = serialized_output.item()
                else:
                    self._raw_data = b""
            elif self._datatype == "BF16":
                serialized_output = serialize_bf16_tensor(input_tensor)
                self._raw_data = serialized_output.tobytes()
            else:
                self._raw_data = input_tensor.tobytes()
            self._parameters["binary_data_size"] = len(self._raw_data)
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
