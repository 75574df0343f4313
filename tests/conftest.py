import sys
from pathlib import Path

import pytest

# Make the repo root importable regardless of how pytest is invoked.
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU")


@pytest.fixture(scope="module")
def http_fixture_server():
    """A KServe-v2 HTTP server with the canonical fixture models on an
    ephemeral port. Yields (host, port, core)."""
    from client_amd.server import (
        AddSubModel,
        HttpServer,
        IdentityModel,
        InferenceCore,
        RepeatModel,
        SequenceModel,
    )

    core = InferenceCore()
    core.add_model(IdentityModel("identity_fp32", "FP32"))
    core.add_model(IdentityModel("identity_bf16", "BF16"))
    core.add_model(IdentityModel("identity_bytes", "BYTES"))
    core.add_model(IdentityModel("identity_int8", "INT8"))
    core.add_model(AddSubModel("simple", "INT32", (-1, 16)))
    core.add_model(AddSubModel("simple_string", "BYTES", (-1, 16)))
    core.add_model(SequenceModel())
    core.add_model(RepeatModel())
    server = HttpServer(core, host="127.0.0.1", port=0)
    stop = server.serve_forever_in_thread()
    yield "127.0.0.1", server.port, core
    stop()


def _make_fixture_core():
    from client_amd.server import (
        AddSubModel,
        IdentityModel,
        InferenceCore,
        RepeatModel,
        SequenceModel,
    )

    core = InferenceCore()
    core.add_model(IdentityModel("identity_fp32", "FP32"))
    core.add_model(IdentityModel("identity_bf16", "BF16"))
    core.add_model(IdentityModel("identity_bytes", "BYTES"))
    core.add_model(IdentityModel("identity_int8", "INT8"))
    core.add_model(AddSubModel("simple", "INT32", (-1, 16)))
    core.add_model(AddSubModel("simple_string", "BYTES", (-1, 16)))
    core.add_model(SequenceModel())
    core.add_model(RepeatModel())
    return core


@pytest.fixture(scope="module")
def grpc_fixture_server():
    """A KServe-v2 gRPC server with the fixture models on an ephemeral
    port. Yields (host, port, core)."""
    from client_amd.server.grpc_server import GrpcServer

    core = _make_fixture_core()
    server = GrpcServer(core, host="127.0.0.1", port=0)
    server.start()
    yield "127.0.0.1", server.port, core
    server.stop(grace=1)
