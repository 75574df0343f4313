from client_amd.http import *  # noqa: F401,F403
from client_amd.http import (  # noqa: F401
    InferenceServerClient, InferAsyncRequest, InferInput, InferResult,
    InferRequestedOutput, InferenceServerException,
)
