"""Deprecated: use tritonclient.http (reference shim:
tritonhttpclient/__init__.py:28-40)."""
import warnings

warnings.warn(
    "The package `tritonhttpclient` is deprecated. Use `tritonclient.http`.",
    DeprecationWarning,
)
from tritonclient.http import *  # noqa: F401,F403
from tritonclient.http import InferenceServerClient  # noqa: F401
