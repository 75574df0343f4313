"""HTTP InferRequestedOutput (reference: tritonclient/http/_requested_output.py)."""


class InferRequestedOutput:
    def __init__(self, name, binary_data=True, class_count=0):
        self._name = name
        self._parameters = {}
        if class_count != 0:
            self._parameters["classification"] = class_count
        self._binary = binary_data
        self._parameters["binary_data"] = binary_data

    def name(self):
        return self._name

    def set_shared_memory(self, region_name, byte_size, offset=0):
        """Receive this output into a registered shared-memory region."""
        if "classification" in self._parameters:
            raise ValueError("shared memory can't be set on classification output")
        if self._binary:
            self._parameters["binary_data"] = False
        self._parameters["shared_memory_region"] = region_name
        self._parameters["shared_memory_byte_size"] = byte_size
        if offset != 0:
            self._parameters["shared_memory_offset"] = offset

    def unset_shared_memory(self):
        """Clear the shared-memory binding, restoring the binary_data flag."""
        self._parameters["binary_data"] = self._binary
        self._parameters.pop("shared_memory_region", None)
        self._parameters.pop("shared_memory_byte_size", None)
        self._parameters.pop("shared_memory_offset", None)

    def _get_tensor(self):
        tensor = {"name": self._name}
        if self._parameters:
            tensor["parameters"] = self._parameters
        return tensor
