"""client_amd.ops — CDNA4 HIP kernels and HIP runtime surface.

``hip_runtime`` is the loud-failure import point: on a machine with an
AMD GPU the native extension MUST load (no silent CPU fallback — the
HIP path is the product); on CPU-only machines importing this package
is fine and only using a GPU function raises.
"""

import os as _os

# must be in the environment before the HIP runtime initializes:
# the host driver only supports dmabuf IPC (hipIpc* fails EINVAL
# otherwise)
_os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")

from . import build as _build


class _HipRuntimeProxy:
    """Lazy loader for the _hip_c extension with loud failure on GPU."""

    def __init__(self):
        self._mod = None
        self._err = None

    def _load(self):
        if self._mod is not None:
            return self._mod
        # One process, ONE HIP runtime: torch bundles its own
        # libamdhip64 with the same SONAME as /opt/rocm's. Importing
        # torch first makes the dynamic linker bind _hip_c to torch's
        # already-loaded copy; loading ours first breaks torch.cuda in
        # this process and cross-process IPC against torch processes.
        try:
            import torch  # noqa: F401
        except ImportError:
            pass
        try:
            from . import _hip_c  # noqa

            self._mod = _hip_c
            return self._mod
        except ImportError as e:
            self._err = e
            # Attempt an in-tree build once (hipcc cross-compiles anywhere).
            try:
                _build.build()
                import importlib

                from . import _hip_c  # noqa

                self._mod = _hip_c
                return self._mod
            except Exception as be:
                raise RuntimeError(
                    "client_amd native HIP extension (_hip_c.so) is not "
                    "available and in-tree build failed. On a GPU machine "
                    "this is fatal — there is no CPU fallback for the HIP "
                    f"data plane. Import error: {e}; build error: {be}"
                ) from be

    def __getattr__(self, name):
        return getattr(self._load(), name)


hip_runtime = _HipRuntimeProxy()


def gpu_available():
    """True if the extension loads and at least one HIP device exists."""
    try:
        return hip_runtime.device_count() > 0
    except Exception:
        return False
