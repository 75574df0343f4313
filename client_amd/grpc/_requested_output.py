"""gRPC InferRequestedOutput (reference: tritonclient/grpc/_requested_output.py)."""

from ..utils import raise_error
from ._proto import service_pb2


class InferRequestedOutput:
    def __init__(self, name, class_count=0):
        self._output = service_pb2.ModelInferRequest.InferRequestedOutputTensor()
        self._output.name = name
        if class_count != 0:
            self._output.parameters["classification"].int64_param = class_count

    def name(self):
        return self._output.name

    def set_shared_memory(self, region_name, byte_size, offset=0):
        if "classification" in self._output.parameters:
            raise_error("shared memory can't be set on classification output")
        self._output.parameters["shared_memory_region"].string_param = region_name
        self._output.parameters["shared_memory_byte_size"].int64_param = byte_size
        if offset != 0:
            self._output.parameters["shared_memory_offset"].int64_param = offset

    def unset_shared_memory(self):
        self._output.parameters.pop("shared_memory_region", None)
        self._output.parameters.pop("shared_memory_byte_size", None)
        self._output.parameters.pop("shared_memory_offset", None)

    def _get_tensor(self):
        return self._output
