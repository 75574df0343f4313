"""Deprecated: use tritonclient.grpc."""
import warnings

warnings.warn(
    "The package `tritongrpcclient` is deprecated. Use `tritonclient.grpc`.",
    DeprecationWarning,
)
from tritonclient.grpc import *  # noqa: F401,F403
from tritonclient.grpc import InferenceServerClient  # noqa: F401
