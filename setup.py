#!/usr/bin/env python3
"""Packaging for the MI355X-native Triton client stack.

Mirrors the reference wheel layout (src/python/library/setup.py:44-77):
the ``tritonclient`` import name with http/grpc/all extras; the ``cuda``
extra is kept as an alias of ``hip`` for compatibility. Building with
``python setup.py build_ext --inplace`` compiles the gfx950 HIP
extension in-tree via hipcc.
"""

import os
import subprocess
import sys

from setuptools import Command, find_packages, setup

REPO = os.path.dirname(os.path.abspath(__file__))


class BuildHipExt(Command):
    description = "build the _hip_c extension for gfx950 (hipcc)"
    user_options = [("inplace", "i", "build in the source tree (default)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        sys.path.insert(0, REPO)
        from client_amd.ops import build as ops_build

        ops_build.build(force=False)


platform_package_data = ["_hip_c.so"]

setup(
    name="client_amd",
    version="0.1.0",
    description=(
        "MI355X-native Triton Inference Server client stack: KServe-v2 "
        "HTTP/gRPC clients, HIP-IPC shared memory, CDNA4 pack kernels, "
        "perf_analyzer-class load generator"
    ),
    license="BSD",
    packages=find_packages(
        include=[
            "client_amd*",
            "tritonclient*",
            "tritonhttpclient",
            "tritongrpcclient",
            "tritonclientutils",
            "tritonshmutils",
        ]
    ),
    package_data={"client_amd.ops": platform_package_data},
    python_requires=">=3.8",
    install_requires=["numpy>=1.20"],
    extras_require={
        "http": ["aiohttp>=3.8"],
        "grpc": ["grpcio>=1.50", "protobuf>=4.0"],
        # hip == cuda extra: the GPU path is HIP-IPC; the name 'cuda' is
        # kept so tritonclient[cuda] installs keep working.
        "hip": [],
        "cuda": [],
        "all": ["aiohttp>=3.8", "grpcio>=1.50", "protobuf>=4.0"],
    },
    cmdclass={"build_hip": BuildHipExt},
)
