"""client_amd.grpc — KServe-v2 gRPC client.

Mirrors tritonclient.grpc's public surface
(reference: /root/reference/src/python/library/tritonclient/grpc/__init__.py).
"""

from ..utils import InferenceServerException
from ._client import (
    CallContext,
    InferenceServerClient,
    KeepAliveOptions,
    MAX_GRPC_MESSAGE_SIZE,
)
from ._infer_input import InferInput
from ._infer_result import InferResult
from . import model_config_pb2
from ._proto import service_pb2
from ._requested_output import InferRequestedOutput

__all__ = [
    "InferenceServerClient",
    "InferInput",
    "InferResult",
    "InferRequestedOutput",
    "InferenceServerException",
    "KeepAliveOptions",
    "CallContext",
    "MAX_GRPC_MESSAGE_SIZE",
    "service_pb2",
]
