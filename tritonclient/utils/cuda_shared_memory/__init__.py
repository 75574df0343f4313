"""On this stack cuda_shared_memory IS the HIP-IPC implementation —
same API, hipMalloc/hipIpc* underneath (no CUDA anywhere)."""
from client_amd.utils.hip_shared_memory import *  # noqa: F401,F403
from client_amd.utils.hip_shared_memory import (  # noqa: F401
    CudaSharedMemoryException, CudaSharedMemoryRegion,
    create_shared_memory_region, get_raw_handle, set_shared_memory_region,
    set_shared_memory_region_from_dlpack, get_contents_as_numpy,
    as_shared_memory_tensor, allocated_shared_memory_regions,
    destroy_shared_memory_region,
)
