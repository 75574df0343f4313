"""gRPC InferResult (reference: tritonclient/grpc/_infer_result.py).

Zero-copy views into ModelInferResponse.raw_output_contents; falls back
to typed ``contents`` fields when the server used them. Also surfaces
the decoupled-stream final/null-response parameters
(reference grpc_client.cc:414-419, common.h:534-540).
"""

import numpy as np

from ..utils import (
    deserialize_bf16_tensor,
    deserialize_bytes_tensor,
    triton_to_np_dtype,
)


class InferResult:
    def __init__(self, result):
        self._result = result

    @staticmethod
    def _uses_raw_slot(output):
        """True if this output consumes an entry of raw_output_contents:
        it carries neither typed contents nor a shared-memory reference."""
        if "shared_memory_region" in output.parameters:
            return False
        c = output.contents
        return not (
            len(c.bool_contents) or len(c.int_contents) or len(c.int64_contents)
            or len(c.uint_contents) or len(c.uint64_contents)
            or len(c.fp32_contents) or len(c.fp64_contents)
            or len(c.bytes_contents)
        )

    def as_numpy(self, name):
        index = 0
        for output in self._result.outputs:
            if output.name == name:
                if "shared_memory_region" in output.parameters:
                    # data lives in the client's shared-memory region
                    return None
                shape = list(output.shape)
                datatype = output.datatype
                if self._uses_raw_slot(output) and index < len(
                    self._result.raw_output_contents
                ):
                    if datatype == "BYTES":
                        np_array = deserialize_bytes_tensor(
                            self._result.raw_output_contents[index]
                        )
                    elif datatype == "BF16":
                        np_array = deserialize_bf16_tensor(
                            self._result.raw_output_contents[index]
                        )
                    else:
                        np_array = np.frombuffer(
                            self._result.raw_output_contents[index],
                            dtype=triton_to_np_dtype(datatype),
                        )
                elif len(output.contents.bytes_contents) != 0:
                    np_array = np.array(
                        list(output.contents.bytes_contents), dtype=np.object_
                    )
                elif len(output.contents.fp32_contents) != 0:
                    np_array = np.array(output.contents.fp32_contents, dtype=np.float32)
                elif len(output.contents.fp64_contents) != 0:
                    np_array = np.array(output.contents.fp64_contents, dtype=np.float64)
                elif len(output.contents.int_contents) != 0:
                    np_array = np.array(
                        output.contents.int_contents,
                        dtype=triton_to_np_dtype(datatype),
                    )
                elif len(output.contents.int64_contents) != 0:
                    np_array = np.array(output.contents.int64_contents, dtype=np.int64)
                elif len(output.contents.uint_contents) != 0:
                    np_array = np.array(
                        output.contents.uint_contents,
                        dtype=triton_to_np_dtype(datatype),
                    )
                elif len(output.contents.uint64_contents) != 0:
                    np_array = np.array(
                        output.contents.uint64_contents, dtype=np.uint64
                    )
                elif len(output.contents.bool_contents) != 0:
                    np_array = np.array(output.contents.bool_contents, dtype=bool)
                else:
                    np_array = np.empty(0, dtype=triton_to_np_dtype(datatype))
                np_array = np_array.reshape(shape)
                return np_array
            elif self._uses_raw_slot(output):
                index += 1
        return None

    def get_output(self, name, as_json=False):
        for output in self._result.outputs:
            if output.name == name:
                if as_json:
                    from google.protobuf.json_format import MessageToDict

                    return MessageToDict(output, preserving_proto_field_name=True)
                return output
        return None

    def get_response(self, as_json=False):
        if as_json:
            from google.protobuf.json_format import MessageToDict

            return MessageToDict(self._result, preserving_proto_field_name=True)
        return self._result

    # ---- decoupled-stream helpers (reference common.h:534-540) ----
    def is_final_response(self):
        p = self._result.parameters.get("triton_final_response")
        return bool(p.bool_param) if p is not None else None

    def is_null_response(self):
        p = self._result.parameters.get("triton_null_response")
        return bool(p.bool_param) if p is not None else False
