"""In-tree build of the _hip_c extension.

Drives hipcc directly (no hipify, no CUDAExtension translation):
  hipcc --offload-arch=gfx950 -O3 -std=c++17 -shared -fPIC ...
The .so lands next to this file so it ships with the repo snapshot to
the GPU box (JIT caches under ~/.cache do not).
"""

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
SRC = OPS_DIR / "csrc" / "hip_runtime.cpp"
KERNELS = OPS_DIR / "csrc" / "kernels.hip"
OUT = OPS_DIR / "_hip_c.so"

GFX_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950").split(";")[0]


def _pybind11_includes():
    import pybind11

    return [pybind11.get_include()]


HASH_SIDECAR = OPS_DIR / "_hip_c.so.srchash"


def _source_hash():
    import hashlib

    h = hashlib.sha256()
    for p in (SRC, KERNELS):
        h.update(p.read_bytes())
    h.update(GFX_ARCH.encode())
    return h.hexdigest()


def needs_build():
    """True unless the existing .so was provably compiled from the
    current sources: a sidecar file records the source hash at build
    time (an mtime check could silently run a stale committed binary
    on a fresh checkout)."""
    if not OUT.exists() or not HASH_SIDECAR.exists():
        return True
    return HASH_SIDECAR.read_text().strip() != _source_hash()


def build(force=False, verbose=True):
    """Compile the extension for gfx950. Cross-compiles fine on a box
    with no GPU (hipcc only needs the target arch)."""
    if not force and not needs_build():
        return str(OUT)
    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    includes = _pybind11_includes() + [sysconfig.get_paths()["include"]]
    cmd = [
        hipcc,
        f"--offload-arch={GFX_ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-fvisibility=hidden",
        "-Wno-unused-result",
        "-x", "hip",
        str(SRC),
        "-o",
        str(OUT),
    ] + [f"-I{inc}" for inc in includes]
    if verbose:
        print("[client_amd.ops.build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    HASH_SIDECAR.write_text(_source_hash() + "\n")
    return str(OUT)


if __name__ == "__main__":
    build(force="--force" in sys.argv)
