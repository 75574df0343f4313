"""TLS end-to-end: self-signed cert (IP + DNS SANs), HTTPS fixture
server and a TLS gRPC fixture (grpcio secure port, ALPN h2). Covers the
Python sync HTTP client (ssl=True), the Python gRPC client
(ssl_channel_credentials), the C++ HTTP client (HttpSslOptions) and the
C++ gRPC client (SslOptions over the from-scratch h2+OpenSSL path, with
full peer/IP-SAN verification against the self-signed root)."""

import ssl as ssl_mod
import subprocess
from pathlib import Path

import numpy as np
import pytest

import client_amd.http as httpclient


@pytest.fixture(scope="module")
def tls_cert(tmp_path_factory):
    tmp = tmp_path_factory.mktemp("tls")
    cert = tmp / "cert.pem"
    key = tmp / "key.pem"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(key), "-out", str(cert), "-days", "1",
         "-subj", "/CN=localhost",
         "-addext", "subjectAltName=IP:127.0.0.1,DNS:localhost"],
        check=True, capture_output=True,
    )
    return cert, key


def _addsub_core():
    from client_amd.server import AddSubModel, InferenceCore

    core = InferenceCore()
    core.add_model(AddSubModel("simple", "INT32", (-1, 16)))
    return core


@pytest.fixture(scope="module")
def tls_server(tls_cert):
    from client_amd.server import HttpServer

    cert, key = tls_cert
    ctx = ssl_mod.SSLContext(ssl_mod.PROTOCOL_TLS_SERVER)
    ctx.load_cert_chain(str(cert), str(key))
    server = HttpServer(_addsub_core(), host="127.0.0.1", port=0)
    stop = server.serve_forever_in_thread(ssl_context=ctx)
    yield "127.0.0.1", server.port
    stop()


@pytest.fixture(scope="module")
def grpc_tls_server(tls_cert):
    import grpc

    from client_amd.server.grpc_server import GrpcServer

    cert, key = tls_cert
    creds = grpc.ssl_server_credentials(
        [(key.read_bytes(), cert.read_bytes())]
    )
    server = GrpcServer(
        _addsub_core(), host="127.0.0.1", port=0, ssl_credentials=creds
    )
    server.start()
    yield "127.0.0.1", server.port
    server.stop()


def _addsub_io():
    a = np.full((1, 16), 2, dtype=np.int32)
    b = np.full((1, 16), 3, dtype=np.int32)
    return a, b


def test_python_https(tls_server):
    host, port = tls_server
    client = httpclient.InferenceServerClient(
        f"{host}:{port}", ssl=True, insecure=True
    )
    try:
        assert client.is_server_live()
        a, b = _addsub_io()
        inputs = [
            httpclient.InferInput("INPUT0", [1, 16], "INT32"),
            httpclient.InferInput("INPUT1", [1, 16], "INT32"),
        ]
        inputs[0].set_data_from_numpy(a)
        inputs[1].set_data_from_numpy(b)
        result = client.infer("simple", inputs)
        np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), a + b)
    finally:
        client.close()


def test_python_grpc_tls(grpc_tls_server, tls_cert):
    import client_amd.grpc as grpcclient

    host, port = grpc_tls_server
    cert, _ = tls_cert
    client = grpcclient.InferenceServerClient(
        f"{host}:{port}",
        ssl=True,
        root_certificates=str(cert),
        channel_args=[("grpc.ssl_target_name_override", "localhost")],
    )
    try:
        assert client.is_server_live()
        a, b = _addsub_io()
        inputs = [
            grpcclient.InferInput("INPUT0", [1, 16], "INT32"),
            grpcclient.InferInput("INPUT1", [1, 16], "INT32"),
        ]
        inputs[0].set_data_from_numpy(a)
        inputs[1].set_data_from_numpy(b)
        result = client.infer("simple", inputs)
        np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), a + b)
    finally:
        client.close()


def test_cpp_https(tls_server):
    from tests.test_cpp_client import _compile, CPP

    host, port = tls_server
    binary = _compile("tls_smoke", CPP / "tests" / "tls_smoke.cc")
    proc = subprocess.run(
        [str(binary), f"{host}:{port}"], capture_output=True, text=True,
        timeout=60,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "PASS" in proc.stdout


def test_cpp_grpc_tls(grpc_tls_server, tls_cert):
    from tests.test_cpp_client import _compile, CPP

    host, port = grpc_tls_server
    cert, _ = tls_cert
    binary = _compile("grpc_tls_smoke", CPP / "tests" / "grpc_tls_smoke.cc")
    proc = subprocess.run(
        [str(binary), host, str(port), str(cert)],
        capture_output=True, text=True, timeout=60,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "PASS" in proc.stdout


@pytest.fixture(scope="module")
def grpc_mtls_server(tls_cert):
    """Secure port that REQUIRES a client certificate (mTLS); the
    self-signed cert doubles as the client identity and trust root."""
    import grpc

    from client_amd.server.grpc_server import GrpcServer

    cert, key = tls_cert
    creds = grpc.ssl_server_credentials(
        [(key.read_bytes(), cert.read_bytes())],
        root_certificates=cert.read_bytes(),
        require_client_auth=True,
    )
    server = GrpcServer(
        _addsub_core(), host="127.0.0.1", port=0, ssl_credentials=creds
    )
    server.start()
    yield "127.0.0.1", server.port
    server.stop()


def test_python_grpc_mtls(grpc_mtls_server, tls_cert):
    import client_amd.grpc as grpcclient

    host, port = grpc_mtls_server
    cert, key = tls_cert
    client = grpcclient.InferenceServerClient(
        f"{host}:{port}",
        ssl=True,
        root_certificates=str(cert),
        private_key=str(key),
        certificate_chain=str(cert),
        channel_args=[("grpc.ssl_target_name_override", "localhost")],
    )
    try:
        assert client.is_server_live()
    finally:
        client.close()


def test_cpp_grpc_mtls(grpc_mtls_server, tls_cert):
    from tests.test_cpp_client import _compile, CPP

    host, port = grpc_mtls_server
    cert, key = tls_cert
    binary = _compile("grpc_tls_smoke", CPP / "tests" / "grpc_tls_smoke.cc")
    proc = subprocess.run(
        [str(binary), host, str(port), str(cert), str(key), str(cert)],
        capture_output=True, text=True, timeout=60,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "PASS" in proc.stdout


def test_cpp_grpc_tls_rejects_without_client_cert(grpc_mtls_server, tls_cert):
    """The mTLS server must refuse a client that presents no cert."""
    from tests.test_cpp_client import _compile, CPP

    host, port = grpc_mtls_server
    cert, _ = tls_cert
    binary = _compile("grpc_tls_smoke", CPP / "tests" / "grpc_tls_smoke.cc")
    proc = subprocess.run(
        [str(binary), host, str(port), str(cert)],
        capture_output=True, text=True, timeout=60,
    )
    assert proc.returncode != 0


def test_cpp_grpc_ssl_example(grpc_tls_server, tls_cert):
    from tests.test_cpp_client import _compile, CPP

    host, port = grpc_tls_server
    cert, _ = tls_cert
    binary = _compile("simple_grpc_ssl_infer_client",
                      CPP / "examples" / "simple_grpc_ssl_infer_client.cc")
    proc = subprocess.run(
        [str(binary), "-u", f"{host}:{port}", "-ssl",
         "--root-certificates", str(cert)],
        capture_output=True, text=True, timeout=60,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "PASS" in proc.stdout


def test_python_aio_https(tls_server, tls_cert):
    """asyncio HTTP client over TLS with full verification against the
    self-signed root (IP SAN)."""
    import asyncio

    import client_amd.http.aio as aiohttpclient

    host, port = tls_server
    cert, _ = tls_cert

    async def run():
        ctx = ssl_mod.create_default_context(cafile=str(cert))
        client = aiohttpclient.InferenceServerClient(
            f"{host}:{port}", ssl=True, ssl_context=ctx
        )
        try:
            assert await client.is_server_live()
            a, b = _addsub_io()
            inputs = [
                aiohttpclient.InferInput("INPUT0", [1, 16], "INT32"),
                aiohttpclient.InferInput("INPUT1", [1, 16], "INT32"),
            ]
            inputs[0].set_data_from_numpy(a)
            inputs[1].set_data_from_numpy(b)
            result = await client.infer("simple", inputs)
            np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), a + b)
        finally:
            await client.close()

    asyncio.run(run())


def test_python_aio_grpc_tls(grpc_tls_server, tls_cert):
    import asyncio

    import client_amd.grpc.aio as aiogrpcclient

    host, port = grpc_tls_server
    cert, _ = tls_cert

    async def run():
        client = aiogrpcclient.InferenceServerClient(
            f"{host}:{port}",
            ssl=True,
            root_certificates=str(cert),
            channel_args=[("grpc.ssl_target_name_override", "localhost")],
        )
        try:
            assert await client.is_server_live()
        finally:
            await client.close()

    asyncio.run(run())
