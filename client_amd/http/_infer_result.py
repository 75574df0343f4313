"""HTTP InferResult (reference: tritonclient/http/_infer_result.py).

Parses a KServe-v2 infer response: optional gzip/deflate decompression,
split of the JSON header from the trailing binary buffer at
``Inference-Header-Content-Length``, output-name -> byte-offset map, and
``as_numpy`` with BYTES/BF16 deserialization.
"""

import gzip
import json
import zlib

import numpy as np

from ..utils import (
    deserialize_bf16_tensor,
    deserialize_bytes_tensor,
    raise_error,
    triton_to_np_dtype,
)


class InferResult:
    def __init__(self, response, verbose):
        """``response`` is an object with ``.headers`` (case-insensitive
        mapping) and ``.read()`` returning the raw body bytes."""
        # keep response headers reachable: the ORCA per-response load
        # metrics ride in `endpoint-load-metrics[-format]`
        # (reference README.md:352-366)
        self._headers = response.headers
        header_length_str = response.headers.get(
            "Inference-Header-Content-Length", None
        )
        content_encoding = response.headers.get("Content-Encoding", None)
        content = response.read()
        if content_encoding is not None:
            if content_encoding == "gzip":
                content = gzip.decompress(content)
            elif content_encoding == "deflate":
                content = zlib.decompress(content)
        if header_length_str is None:
            content = content.decode("utf-8") if isinstance(content, bytes) else content
            if verbose:
                print(content)
            self._result = json.loads(content)
            self._output_name_to_buffer_map = {}
            self._buffer = None
        else:
            header_length = int(header_length_str)
            self._result = json.loads(content[:header_length])
            if verbose:
                print(self._result)
            self._output_name_to_buffer_map = {}
            self._buffer = content[header_length:]
            offset = 0
            if "outputs" in self._result:
                for output in self._result["outputs"]:
                    parameters = output.get("parameters", {})
                    if "binary_data_size" in parameters:
                        byte_size = parameters["binary_data_size"]
                        self._output_name_to_buffer_map[output["name"]] = offset
                        offset += byte_size

    @classmethod
    def from_response_body(
        cls, response_body, verbose=False, header_length=None, content_encoding=None
    ):
        """Build an InferResult from raw body bytes (out-of-band use;
        reference _infer_result.py:108-155)."""

        class _FakeResponse:
            def __init__(self, body, headers):
                self._body = body
                self.headers = headers

            def read(self):
                return self._body

        headers = {}
        if header_length is not None:
            headers["Inference-Header-Content-Length"] = str(header_length)
        if content_encoding is not None:
            headers["Content-Encoding"] = content_encoding
        return cls(_FakeResponse(response_body, headers), verbose)

    def as_numpy(self, name):
        """Return output tensor ``name`` as a numpy array (None if absent)."""
        if self._result.get("outputs") is not None:
            for output in self._result["outputs"]:
                if output["name"] == name:
                    datatype = output["datatype"]
                    has_binary_data = False
                    parameters = output.get("parameters", {})
                    if "binary_data_size" in parameters:
                        has_binary_data = True
                        byte_size = parameters["binary_data_size"]
                        start = self._output_name_to_buffer_map[name]
                        data_buffer = self._buffer[start : start + byte_size]
                        if datatype == "BYTES":
                            np_array = deserialize_bytes_tensor(data_buffer)
                        elif datatype == "BF16":
                            np_array = deserialize_bf16_tensor(data_buffer)
                        else:
                            np_array = np.frombuffer(
                                data_buffer, dtype=triton_to_np_dtype(datatype)
                            )
                    if not has_binary_data:
                        if "data" not in output:
                            return None
                        if datatype == "BYTES":
                            np_array = np.array(
                                [
                                    val.encode("utf-8") if isinstance(val, str) else val
                                    for val in output["data"]
                                ],
                                dtype=np.object_,
                            )
                        elif datatype == "BF16":
                            raise_error(
                                "BF16 outputs must be requested as binary data"
                            )
                        else:
                            np_array = np.array(
                                output["data"], dtype=triton_to_np_dtype(datatype)
                            )
                    np_array = np_array.reshape(output["shape"])
                    return np_array
        return None

    def get_output(self, name):
        """Return the JSON dict describing output ``name`` (or None)."""
        for output in self._result.get("outputs", []):
            if output["name"] == name:
                return output
        return None

    def get_response(self):
        """The full parsed JSON response."""
        return self._result

    def get_response_header(self, name, default=None):
        """A raw response header (e.g. the ORCA
        ``endpoint-load-metrics`` reported per response)."""
        try:
            return self._headers.get(name, default)
        except AttributeError:
            return default
