"""Deprecated: use tritonclient.utils.shared_memory / cuda_shared_memory."""
import warnings

warnings.warn(
    "The package `tritonshmutils` is deprecated. Use "
    "`tritonclient.utils.shared_memory`.",
    DeprecationWarning,
)
import tritonclient.utils.shared_memory as shared_memory  # noqa: F401
import tritonclient.utils.cuda_shared_memory as cuda_shared_memory  # noqa: F401
