"""Deprecated: use tritonclient.utils."""
import warnings

warnings.warn(
    "The package `tritonclientutils` is deprecated. Use `tritonclient.utils`.",
    DeprecationWarning,
)
from tritonclient.utils import *  # noqa: F401,F403
