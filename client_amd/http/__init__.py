"""client_amd.http — KServe-v2 HTTP/REST client (sync).

Mirrors tritonclient.http's public surface
(reference: /root/reference/src/python/library/tritonclient/http/__init__.py).
"""

from ..utils import InferenceServerException
from ._client import InferAsyncRequest, InferenceServerClient
from ._infer_input import InferInput
from ._infer_result import InferResult
from ._requested_output import InferRequestedOutput

__all__ = [
    "InferenceServerClient",
    "InferAsyncRequest",
    "InferInput",
    "InferResult",
    "InferRequestedOutput",
    "InferenceServerException",
]
