from client_amd.grpc import *  # noqa: F401,F403
from client_amd.grpc import (  # noqa: F401
    InferenceServerClient, InferInput, InferResult, InferRequestedOutput,
    InferenceServerException, KeepAliveOptions, CallContext, service_pb2,
    model_config_pb2,
)
