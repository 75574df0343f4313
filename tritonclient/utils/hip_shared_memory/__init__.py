from client_amd.utils.hip_shared_memory import *  # noqa: F401,F403
