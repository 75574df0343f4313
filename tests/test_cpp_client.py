"""C++ client library tests: compile with g++ (cached) and run the
standalone cc_client_test + examples against the Python fixture server
(SURVEY.md §4 tier 2 — the reference's cc_client_test model)."""

import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
CPP = REPO / "src" / "cpp"
BUILD = CPP / "build_test"


def _compile(name, main_src, grpc=False):
    BUILD.mkdir(exist_ok=True)
    out = BUILD / name
    base = ["common.cc", "json.cc", "shm_utils.cc"]
    base += (["h2.cc", "kserve_pb.cc", "grpc_client.cc"] if grpc
             else ["http_client.cc"])
    base += ["hip_shm.cc"]
    srcs = [CPP / s for s in base] + [main_src]
    newest = max(p.stat().st_mtime for p in srcs + [CPP / "include" /
                                                    "client_amd" / "common.h"])
    if out.exists() and out.stat().st_mtime > newest:
        return out
    cmd = ["g++", "-std=c++17", "-O1", f"-I{CPP}/include", "-Wall",
           *map(str, srcs), "-o", str(out), "-lpthread", "-lrt", "-lz"]
    if grpc:
        cmd.append("-l:libnghttp2.so.14")
    subprocess.run(cmd, check=True, capture_output=True, text=True)
    return out


@pytest.fixture(scope="module")
def cc_binaries():
    try:
        test_bin = _compile("cc_client_test", CPP / "tests" / "cc_client_test.cc")
        example_bin = _compile(
            "simple_http_infer_client",
            CPP / "examples" / "simple_http_infer_client.cc")
        perf_bin = _compile("perf_client", CPP / "examples" / "perf_client.cc")
    except subprocess.CalledProcessError as e:
        pytest.fail(f"C++ compile failed:\n{e.stderr}")
    return test_bin, example_bin, perf_bin


def test_cc_client(cc_binaries, http_fixture_server):
    host, port, _ = http_fixture_server
    test_bin, _, _ = cc_binaries
    proc = subprocess.run(
        [str(test_bin), f"{host}:{port}"], capture_output=True, text=True,
        timeout=120,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "ALL PASSED" in proc.stdout


def test_cc_example(cc_binaries, http_fixture_server):
    host, port, _ = http_fixture_server
    _, example_bin, _ = cc_binaries
    proc = subprocess.run(
        [str(example_bin), "-u", f"{host}:{port}"], capture_output=True,
        text=True, timeout=60,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "PASS" in proc.stdout


@pytest.fixture(scope="module")
def cc_grpc_binaries():
    try:
        test_bin = _compile("cc_grpc_test", CPP / "tests" / "cc_grpc_test.cc",
                            grpc=True)
        example_bin = _compile(
            "simple_grpc_infer_client",
            CPP / "examples" / "simple_grpc_infer_client.cc", grpc=True)
    except subprocess.CalledProcessError as e:
        pytest.fail(f"C++ gRPC compile failed:\n{e.stderr}")
    return test_bin, example_bin


def test_cc_grpc_client(cc_grpc_binaries, grpc_fixture_server):
    """The from-scratch HTTP/2 + hand-encoded protobuf gRPC client
    against the grpcio fixture server."""
    host, port, _ = grpc_fixture_server
    test_bin, _ = cc_grpc_binaries
    proc = subprocess.run(
        [str(test_bin), f"{host}:{port}"], capture_output=True, text=True,
        timeout=120,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "ALL PASSED" in proc.stdout


def test_cc_grpc_example(cc_grpc_binaries, grpc_fixture_server):
    host, port, _ = grpc_fixture_server
    _, example_bin = cc_grpc_binaries
    proc = subprocess.run(
        [str(example_bin), "-u", f"{host}:{port}"], capture_output=True,
        text=True, timeout=60,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "PASS" in proc.stdout


def test_cc_perf_client(cc_binaries, http_fixture_server):
    host, port, _ = http_fixture_server
    _, _, perf_bin = cc_binaries
    proc = subprocess.run(
        [str(perf_bin), "-u", f"{host}:{port}", "-m", "simple",
         "--concurrency-range", "2:2:1", "--measurement-interval", "0.3",
         "--max-windows", "1"],
        capture_output=True, text=True, timeout=60,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "infer/sec" in proc.stdout
    assert "errors: 0" in proc.stdout


def test_cc_perf_client_shm(cc_binaries, http_fixture_server):
    """C++ load generator in system-shm I/O mode."""
    host, port, _ = http_fixture_server
    _, _, perf_bin = cc_binaries
    proc = subprocess.run(
        [str(perf_bin), "-u", f"{host}:{port}", "-m", "simple", "--shm",
         "--concurrency-range", "2:2:1", "--measurement-interval", "0.3",
         "--max-windows", "1"],
        capture_output=True, text=True, timeout=60,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "errors: 0" in proc.stdout
