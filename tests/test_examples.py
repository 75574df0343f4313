"""Run the example scripts against the fixture servers (keeps the
example matrix green — the reference's examples are its de-facto
integration suite)."""

import os
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
EXAMPLES = REPO / "examples" / "python"

HTTP_EXAMPLES = [
    "simple_http_infer_client.py",
    "simple_http_async_infer_client.py",
    "simple_http_string_infer_client.py",
    "simple_http_health_metadata.py",
    "simple_http_model_control.py",
    "simple_http_shm_client.py",
    "simple_http_shm_string_client.py",
    "simple_http_sequence_sync_infer_client.py",
    "simple_http_aio_infer_client.py",
    "reuse_infer_objects_client.py",
    "builder_infer_client.py",
    "rotating_endpoint_client.py",
]

GRPC_EXAMPLES = [
    "simple_grpc_infer_client.py",
    "simple_grpc_async_infer_client.py",
    "simple_grpc_sequence_stream_infer_client.py",
    "simple_grpc_custom_repeat.py",
    "simple_grpc_health_metadata.py",
    "simple_grpc_model_control.py",
    "simple_grpc_keepalive_client.py",
    "simple_grpc_custom_args_client.py",
    "simple_grpc_aio_infer_client.py",
    "simple_grpc_string_infer_client.py",
    "simple_grpc_shm_client.py",
    "simple_grpc_shm_string_client.py",
    "simple_grpc_sequence_sync_infer_client.py",
    "simple_grpc_aio_sequence_stream_infer_client.py",
    "grpc_client.py",
    "grpc_explicit_int_content_client.py",
    "grpc_explicit_int8_content_client.py",
    "grpc_explicit_byte_content_client.py",
]


def _run(script, url):
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO) + os.pathsep + env.get("PYTHONPATH", "")
    proc = subprocess.run(
        [sys.executable, str(EXAMPLES / script), "-u", url],
        capture_output=True, text=True, timeout=120, env=env,
    )
    assert proc.returncode == 0, f"{script}:\n{proc.stdout}\n{proc.stderr}"
    assert "PASS" in proc.stdout, proc.stdout


@pytest.mark.parametrize("script", HTTP_EXAMPLES)
def test_http_example(script, http_fixture_server):
    host, port, _ = http_fixture_server
    _run(script, f"{host}:{port}")


@pytest.mark.parametrize("script", GRPC_EXAMPLES)
def test_grpc_example(script, grpc_fixture_server):
    host, port, _ = grpc_fixture_server
    _run(script, f"{host}:{port}")


def test_memory_growth(http_fixture_server):
    host, port, _ = http_fixture_server
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO) + os.pathsep + env.get("PYTHONPATH", "")
    proc = subprocess.run(
        [sys.executable, str(EXAMPLES / "memory_growth_test.py"), "-u",
         f"{host}:{port}", "-n", "100"],
        capture_output=True, text=True, timeout=180, env=env,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
