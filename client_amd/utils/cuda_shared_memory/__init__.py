"""Compatibility alias: ``cuda_shared_memory`` -> ``hip_shared_memory``.

Existing tritonclient user code imports
``tritonclient.utils.cuda_shared_memory``; on this stack that name is
the HIP-IPC implementation (same function names/signatures, hip runtime
underneath). There is no CUDA anywhere — the alias exists purely so
user code runs unmodified (SURVEY.md §2.2 row cuda_shared_memory).
"""

from ..hip_shared_memory import *  # noqa: F401,F403
from ..hip_shared_memory import (  # noqa: F401
    CudaSharedMemoryException,
    CudaSharedMemoryRegion,
    allocated_shm_regions,
)
