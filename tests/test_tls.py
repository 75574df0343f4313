"""TLS end-to-end: self-signed cert, HTTPS fixture server, Python sync
client (ssl=True) and the C++ client (HttpSslOptions), verification
relaxed for the self-signed cert."""

import ssl as ssl_mod
import subprocess
from pathlib import Path

import numpy as np
import pytest

import client_amd.http as httpclient


@pytest.fixture(scope="module")
def tls_server(tmp_path_factory):
    tmp = tmp_path_factory.mktemp("tls")
    cert = tmp / "cert.pem"
    key = tmp / "key.pem"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(key), "-out", str(cert), "-days", "1",
         "-subj", "/CN=127.0.0.1"],
        check=True, capture_output=True,
    )
    from client_amd.server import AddSubModel, HttpServer, InferenceCore

    core = InferenceCore()
    core.add_model(AddSubModel("simple", "INT32", (-1, 16)))
    ctx = ssl_mod.SSLContext(ssl_mod.PROTOCOL_TLS_SERVER)
    ctx.load_cert_chain(str(cert), str(key))
    server = HttpServer(core, host="127.0.0.1", port=0)
    stop = server.serve_forever_in_thread(ssl_context=ctx)
    yield "127.0.0.1", server.port
    stop()


def test_python_https(tls_server):
    host, port = tls_server
    client = httpclient.InferenceServerClient(
        f"{host}:{port}", ssl=True, insecure=True
    )
    try:
        assert client.is_server_live()
        a = np.full((1, 16), 2, dtype=np.int32)
        b = np.full((1, 16), 3, dtype=np.int32)
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
