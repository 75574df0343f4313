from client_amd.grpc.aio import *  # noqa: F401,F403
