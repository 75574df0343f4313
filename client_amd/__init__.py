"""client_amd — MI355X-native Triton client stack.

A from-scratch, AMD-first implementation of the capabilities of
triton-inference-server/client (the ``tritonclient`` SDK): KServe-v2
HTTP/REST + gRPC client libraries (Python and C++), a HIP-IPC
shared-memory GPU data plane resident in HBM3E, hand-written CDNA4 HIP
kernels for tensor pack/unpack/cast and image preprocessing, RCCL
fan-out over xGMI to multi-GPU server replicas, and a
perf_analyzer-class load generator.

Subpackages
-----------
- ``client_amd.http`` / ``client_amd.grpc``   — protocol clients (+ .aio)
- ``client_amd.utils``                        — dtype/serialization helpers
- ``client_amd.utils.shared_memory``          — POSIX system shm
- ``client_amd.utils.hip_shared_memory``      — HIP-IPC device shm (alias
  ``cuda_shared_memory`` kept for tritonclient API compatibility)
- ``client_amd.ops``                          — CDNA4 HIP kernels
- ``client_amd.server``                       — MI355X-backed KServe-v2 server
  (test fixture + benchmark target; the reference is client-only)
- ``client_amd.perf``                         — load generator
"""

__version__ = "0.1.0"

from ._builder import InferRequestBuilder
from ._endpoint import FixedEndpoint, MultiEndpointClient, RoundRobinEndpoint

__all__ = [
    "InferRequestBuilder",
    "FixedEndpoint",
    "MultiEndpointClient",
    "RoundRobinEndpoint",
    "__version__",
]
