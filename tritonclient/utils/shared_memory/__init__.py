from client_amd.utils.shared_memory import *  # noqa: F401,F403
from client_amd.utils.shared_memory import (  # noqa: F401
    SharedMemoryException, create_shared_memory_region,
    set_shared_memory_region, get_contents_as_numpy,
    mapped_shared_memory_regions, destroy_shared_memory_region,
)
