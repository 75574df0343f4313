"""Fluent inference-request builder — the ergonomic API surface of the
reference Rust client (reference src/rust/triton-client/src/infer.rs:548
InferRequestBuilder; typed with_data_* setters infer.rs:210-447), which
had no equivalent in the reference C++/Python stacks.

    result = (InferRequestBuilder("resnet50")
              .request_id("r1")
              .input_from_numpy("INPUT0", array)
              .shared_memory_input("INPUT1", "region", 4096)
              .output("OUTPUT0", class_count=3)
              .parameter("priority", 7)
              .infer(client))

The builder is protocol-agnostic: ``infer(client)`` (and ``build(client)``)
inspect the client's module to pick the matching InferInput /
InferRequestedOutput classes, so the same builder drives the HTTP and
gRPC clients.
"""

import numpy as np

from .utils import np_to_triton_dtype, raise_error

__all__ = ["InferRequestBuilder"]


def _io_classes_for(client):
    mod = type(client).__module__
    proto = getattr(client, "protocol", None)  # MultiEndpointClient etc.
    if proto == "http" or ".http" in mod:
        from . import http as pkg
    elif proto == "grpc" or ".grpc" in mod:
        from . import grpc as pkg
    else:
        raise_error(
            f"cannot infer protocol from client type {type(client)!r}"
        )
    return pkg.InferInput, pkg.InferRequestedOutput


class InferRequestBuilder:
    def __init__(self, model_name, model_version=""):
        self._model_name = model_name
        self._model_version = model_version
        self._request_id = None
        self._inputs = []   # (name, kind, payload)
        self._outputs = []  # (name, dict)
        self._parameters = {}
        self._sequence = None  # (id, start, end)
        self._timeout = None

    # ---- request options ----
    def request_id(self, rid):
        self._request_id = rid
        return self

    def parameter(self, key, value):
        self._parameters[key] = value
        return self

    def sequence(self, sequence_id, start=False, end=False):
        self._sequence = (sequence_id, start, end)
        return self

    def timeout(self, client_timeout_s):
        self._timeout = client_timeout_s
        return self

    # ---- inputs (the Rust with_data_* family) ----
    def input_from_numpy(self, name, array, datatype=None):
        """Typed tensor input; datatype defaults from the array dtype
        (Rust InferInput::with_data_* infer.rs:210-447)."""
        if not isinstance(array, np.ndarray):
            array = np.asarray(array)
        dt = datatype or np_to_triton_dtype(array.dtype)
        self._inputs.append((name, "numpy", (array, dt)))
        return self

    def input_bytes(self, name, values):
        """BYTES tensor from a list of bytes/str."""
        arr = np.array(values, dtype=np.object_)
        self._inputs.append((name, "numpy", (arr, "BYTES")))
        return self

    def shared_memory_input(self, name, region, byte_size, shape,
                            datatype, offset=0):
        self._inputs.append(
            (name, "shm", (region, byte_size, shape, datatype, offset))
        )
        return self

    # ---- outputs ----
    def output(self, name, class_count=0, binary_data=None):
        self._outputs.append(
            (name, {"class_count": class_count, "binary_data": binary_data})
        )
        return self

    def shared_memory_output(self, name, region, byte_size, offset=0):
        self._outputs.append(
            (name, {"shm": (region, byte_size, offset)})
        )
        return self

    # ---- build / execute ----
    def build(self, client):
        """Materialize (inputs, outputs, kwargs) for ``client.infer``."""
        input_cls, output_cls = _io_classes_for(client)
        is_http = (getattr(client, "protocol", None) == "http"
                   or ".http" in type(client).__module__)
        inputs = []
        for name, kind, payload in self._inputs:
            if kind == "numpy":
                array, dt = payload
                inp = input_cls(name, list(array.shape), dt)
                inp.set_data_from_numpy(array)
            else:
                region, byte_size, shape, dt, offset = payload
                inp = input_cls(name, list(shape), dt)
                inp.set_shared_memory(region, byte_size, offset=offset)
            inputs.append(inp)
        outputs = []
        for name, spec in self._outputs:
            if "shm" in spec:
                out = output_cls(name)
                region, byte_size, offset = spec["shm"]
                out.set_shared_memory(region, byte_size, offset=offset)
            else:
                kwargs = {"class_count": spec["class_count"]}
                if is_http and spec["binary_data"] is not None:
                    kwargs["binary_data"] = spec["binary_data"]
                out = output_cls(name, **kwargs)
            outputs.append(out)
        call_kwargs = {"model_version": self._model_version}
        if self._request_id is not None:
            call_kwargs["request_id"] = self._request_id
        if self._parameters:
            call_kwargs["parameters"] = dict(self._parameters)
        if self._sequence is not None:
            sid, start, end = self._sequence
            call_kwargs.update(sequence_id=sid, sequence_start=start,
                               sequence_end=end)
        if self._timeout is not None:
            key = "timeout" if is_http else "client_timeout"
            call_kwargs[key] = self._timeout
        return inputs, outputs, call_kwargs

    def infer(self, client, **extra_kwargs):
        """Build and execute against either protocol's client."""
        inputs, outputs, kwargs = self.build(client)
        kwargs.update(extra_kwargs)
        return client.infer(self._model_name, inputs,
                            outputs=outputs or None, **kwargs)

    def async_infer(self, client, callback=None, **extra_kwargs):
        inputs, outputs, kwargs = self.build(client)
        kwargs.update(extra_kwargs)
        if callback is not None:
            kwargs["callback"] = callback
        return client.async_infer(self._model_name, inputs,
                                  outputs=outputs or None, **kwargs)
