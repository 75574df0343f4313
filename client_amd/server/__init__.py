"""client_amd.server — MI355X-backed KServe-v2 server.

The reference repo is a client SDK; this server exists so the client
stack is testable offline (SURVEY.md §4 "fake-server fixture") and so
the benchmark has a real MI355X-backed target. Models run
via PyTorch-ROCm in bf16; HIP-IPC shared-memory regions registered by
clients are opened with hipIpcOpenMemHandle and consumed as device
tensors resident in HBM3E.
"""

from .core import InferenceCore, InferenceError, ShmRegistry
from .http_server import HttpServer
from .models import (
    AddSubModel,
    EnsembleModel,
    GenerateModel,
    IdentityModel,
    Model,
    RepeatModel,
    PreprocessModel,
    SequenceModel,
    TorchModel,
)

__all__ = [
    "InferenceCore",
    "InferenceError",
    "ShmRegistry",
    "HttpServer",
    "Model",
    "GenerateModel",
    "EnsembleModel",
    "PreprocessModel",
    "IdentityModel",
    "AddSubModel",
    "SequenceModel",
    "RepeatModel",
    "TorchModel",
]
