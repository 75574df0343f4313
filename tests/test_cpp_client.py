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


_LIB_SRCS = ["common.cc", "json.cc", "shm_utils.cc", "http_client.cc",
             "h2.cc", "kserve_pb.cc", "grpc_client.cc", "hip_shm.cc"]


def _headers_mtime():
    return max(p.stat().st_mtime
               for p in (CPP / "include" / "client_amd").glob("*.h"))


def _object(src_name):
    """Compile one library source to a cached .o."""
    BUILD.mkdir(exist_ok=True)
    src = CPP / src_name
    obj = BUILD / (src_name + ".o")
    if obj.exists() and obj.stat().st_mtime > max(src.stat().st_mtime,
                                                 _headers_mtime()):
        return obj
    subprocess.run(
        ["g++", "-std=c++17", "-O1", f"-I{CPP}/include", "-Wall", "-c",
         str(src), "-o", str(obj)],
        check=True, capture_output=True, text=True)
    return obj


def _compile(name, main_src, grpc=False):
    BUILD.mkdir(exist_ok=True)
    out = BUILD / name
    objs = [_object(s) for s in _LIB_SRCS]
    newest = max([main_src.stat().st_mtime, _headers_mtime()]
                 + [o.stat().st_mtime for o in objs])
    if out.exists() and out.stat().st_mtime > newest:
        return out
    cmd = ["g++", "-std=c++17", "-O1", f"-I{CPP}/include", "-Wall",
           str(main_src), *map(str, objs), "-o", str(out), "-lpthread",
           "-lrt", "-lz", "-lssl", "-lcrypto", "-l:libnghttp2.so.14"]
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


HTTP_CC_EXAMPLES = [
    "simple_http_async_infer_client.cc",
    "simple_http_string_infer_client.cc",
    "simple_http_shm_client.cc",
    "simple_http_health_metadata.cc",
    "simple_http_model_control.cc",
    "simple_http_sequence_sync_infer_client.cc",
    "reuse_infer_objects_client.cc",
]

GRPC_CC_EXAMPLES = [
    "simple_grpc_health_metadata.cc",
    "simple_grpc_model_control.cc",
    "simple_grpc_string_infer_client.cc",
    "simple_grpc_async_infer_client.cc",
    "simple_grpc_sequence_sync_infer_client.cc",
    "simple_grpc_shm_client.cc",
    "simple_grpc_custom_args_client.cc",
    "simple_grpc_custom_repeat.cc",
    "simple_grpc_keepalive_client.cc",
]


@pytest.mark.parametrize("src", GRPC_CC_EXAMPLES)
def test_cc_grpc_examples(src, grpc_fixture_server):
    host, port, _ = grpc_fixture_server
    binary = _compile(src[:-3], CPP / "examples" / src)
    proc = subprocess.run(
        [str(binary), "-u", f"{host}:{port}"], capture_output=True, text=True,
        timeout=60,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "PASS" in proc.stdout


@pytest.mark.parametrize("src", HTTP_CC_EXAMPLES)
def test_cc_http_examples(src, http_fixture_server):
    host, port, _ = http_fixture_server
    binary = _compile(src[:-3], CPP / "examples" / src)
    proc = subprocess.run(
        [str(binary), "-u", f"{host}:{port}"], capture_output=True, text=True,
        timeout=60,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "PASS" in proc.stdout


def test_cc_grpc_sequence_stream_example(grpc_fixture_server):
    host, port, _ = grpc_fixture_server
    binary = _compile("simple_grpc_sequence_stream_client",
                      CPP / "examples" / "simple_grpc_sequence_stream_client.cc")
    proc = subprocess.run(
        [str(binary), "-u", f"{host}:{port}"], capture_output=True, text=True,
        timeout=60,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "PASS" in proc.stdout


def _hipcc_example(name, extra_models="simple", with_kernels=False):
    """Compile a C++ example with hipcc (+kernels.hip when needed) and
    run it against a spawned server."""
    import os
    import time

    BUILD.mkdir(exist_ok=True)
    binary = BUILD / name
    srcs = [CPP / s for s in ("common.cc", "json.cc", "shm_utils.cc",
                              "http_client.cc", "hip_shm.cc")]
    if with_kernels:
        srcs.append(REPO / "client_amd" / "ops" / "csrc" / "kernels.hip")
    srcs.append(CPP / "examples" / f"{name}.cc")
    subprocess.run(
        ["/opt/rocm/bin/hipcc", "--offload-arch=gfx950", "-std=c++17", "-O1",
         f"-I{CPP}/include", "-DTRITON_ENABLE_HIP", "-x", "hip",
         *map(str, srcs), "-o", str(binary), "-lpthread", "-lrt", "-lz", "-lssl", "-lcrypto"],
        check=True, capture_output=True, text=True)
    server = subprocess.Popen(
        [sys.executable, "-m", "client_amd.server", "--http-port", "18511",
         "--models", extra_models],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        cwd=REPO,
    )
    try:
        deadline = time.time() + 60
        ready = False
        while time.time() < deadline:
            line = server.stdout.readline()
            if line.startswith("HTTP_READY"):
                ready = True
                break
        assert ready, "server not ready"
        proc = subprocess.run(
            [str(binary), "-u", "127.0.0.1:18511"], capture_output=True,
            text=True, timeout=120,
        )
        assert proc.returncode == 0, proc.stdout + proc.stderr
        assert "PASS" in proc.stdout
    finally:
        server.terminate()
        server.wait(timeout=10)


@pytest.mark.gpu
def test_cc_hipshm_pack_kernel_example_gpu():
    """C++ client + CDNA4 pack/unpack kernels end-to-end (wire-exact
    bf16 via HIP-shm) — the C++ analog of the Python device data plane."""
    from client_amd.ops import gpu_available

    if not gpu_available():
        pytest.skip("no HIP device")
    _hipcc_example("simple_http_hipshm_pack_client",
                   extra_models="identity_bf16", with_kernels=True)


@pytest.mark.gpu
def test_cc_hipshm_example_gpu():
    """Compile the C++ HIP-IPC example with hipcc and run it against an
    out-of-process server — the C++ analog of the cudashm example
    round trip (GPU)."""
    import os
    import time

    from client_amd.ops import gpu_available

    if not gpu_available():
        pytest.skip("no HIP device")
    BUILD.mkdir(exist_ok=True)
    binary = BUILD / "simple_http_hipshm_client"
    srcs = [CPP / s for s in ("common.cc", "json.cc", "shm_utils.cc",
                              "http_client.cc", "hip_shm.cc")]
    srcs.append(CPP / "examples" / "simple_http_hipshm_client.cc")
    subprocess.run(
        ["/opt/rocm/bin/hipcc", "--offload-arch=gfx950", "-std=c++17", "-O1",
         f"-I{CPP}/include", "-DTRITON_ENABLE_HIP", *map(str, srcs),
         "-o", str(binary), "-lpthread", "-lrt", "-lz", "-lssl", "-lcrypto"],
        check=True, capture_output=True, text=True)
    server = subprocess.Popen(
        [sys.executable, "-m", "client_amd.server", "--http-port", "18511",
         "--models", "simple"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        cwd=REPO,
    )
    try:
        deadline = time.time() + 60
        ready = False
        while time.time() < deadline:
            line = server.stdout.readline()
            if line.startswith("HTTP_READY"):
                ready = True
                break
        assert ready, "server not ready"
        proc = subprocess.run(
            [str(binary), "-u", "127.0.0.1:18511"], capture_output=True,
            text=True, timeout=120,
        )
        assert proc.returncode == 0, proc.stdout + proc.stderr
        assert "PASS" in proc.stdout
    finally:
        server.terminate()
        server.wait(timeout=10)


@pytest.mark.gpu
def test_cc_grpc_hipshm_example_gpu():
    """gRPC HIP-IPC example (raw 64-byte handle in the proto) compiled
    with hipcc against an out-of-process server — the gRPC analog of
    the cudashm example round trip (GPU)."""
    import time

    from client_amd.ops import gpu_available

    if not gpu_available():
        pytest.skip("no HIP device")
    BUILD.mkdir(exist_ok=True)
    binary = BUILD / "simple_grpc_hipshm_client"
    srcs = [CPP / s for s in ("common.cc", "json.cc", "shm_utils.cc",
                              "http_client.cc", "h2.cc", "kserve_pb.cc",
                              "grpc_client.cc", "hip_shm.cc")]
    srcs.append(CPP / "examples" / "simple_grpc_hipshm_client.cc")
    subprocess.run(
        ["/opt/rocm/bin/hipcc", "--offload-arch=gfx950", "-std=c++17", "-O1",
         f"-I{CPP}/include", "-DTRITON_ENABLE_HIP", "-x", "hip",
         *map(str, srcs), "-o", str(binary), "-lpthread", "-lrt", "-lz",
         "-lssl", "-lcrypto", "-l:libnghttp2.so.14"],
        check=True, capture_output=True, text=True)
    server = subprocess.Popen(
        [sys.executable, "-m", "client_amd.server", "--grpc-port", "18512",
         "--models", "simple"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        cwd=REPO,
    )
    try:
        deadline = time.time() + 60
        ready = False
        while time.time() < deadline:
            line = server.stdout.readline()
            if line.startswith("GRPC_READY"):
                ready = True
                break
        assert ready, "server not ready"
        proc = subprocess.run(
            [str(binary), "-u", "127.0.0.1:18512"], capture_output=True,
            text=True, timeout=120,
        )
        assert proc.returncode == 0, proc.stdout + proc.stderr
        assert "PASS" in proc.stdout
    finally:
        server.terminate()
        server.wait(timeout=10)


def test_cc_image_clients_compile():
    """image_client / ensemble_image_client need a GPU-backed resnet50
    to run; on CPU we verify they build and link."""
    _compile("image_client", CPP / "examples" / "image_client.cc")
    _compile("ensemble_image_client",
             CPP / "examples" / "ensemble_image_client.cc")


def test_cc_keepalive_watchdog_kills_dead_connection():
    """A server that accepts h2 but never ACKs PINGs must be declared
    dead by the keepalive watchdog, failing the in-flight Infer fast
    (grpc keepalive.md semantics)."""
    import socket
    import threading

    binary = _compile("keepalive_timeout_smoke",
                      CPP / "tests" / "keepalive_timeout_smoke.cc")

    srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)
    port = srv.getsockname()[1]
    stop = threading.Event()

    def fake_h2_server():
        try:
            conn, _ = srv.accept()
            conn.settimeout(0.5)
            # empty SETTINGS so the client sees a live h2 peer
            conn.sendall(b"\x00\x00\x00\x04\x00\x00\x00\x00\x00")
            while not stop.is_set():
                try:
                    if conn.recv(65536) == b"":
                        break  # drained; never answer anything
                except socket.timeout:
                    continue
                except OSError:
                    break
            conn.close()
        except Exception:
            pass

    t = threading.Thread(target=fake_h2_server, daemon=True)
    t.start()
    try:
        proc = subprocess.run(
            [str(binary), "127.0.0.1", str(port)], capture_output=True,
            text=True, timeout=30,
        )
        assert proc.returncode == 0, proc.stdout + proc.stderr
        assert "PASS" in proc.stdout
    finally:
        stop.set()
        srv.close()
        t.join(timeout=5)


def test_cc_client_timeout_suite(http_fixture_server, grpc_fixture_server):
    """Reference client_timeout_test analog: every API with microscopic
    deadlines fails cleanly with Deadline Exceeded on both transports."""
    http_host, http_port, _ = http_fixture_server
    grpc_host, grpc_port, _ = grpc_fixture_server
    binary = _compile("client_timeout_test",
                      CPP / "tests" / "client_timeout_test.cc")
    proc = subprocess.run(
        [str(binary), f"{http_host}:{http_port}", f"{grpc_host}:{grpc_port}"],
        capture_output=True, text=True, timeout=120,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "ALL PASSED" in proc.stdout


def test_cc_memory_leak_suite(http_fixture_server, grpc_fixture_server):
    """Reference memory_leak_test analog: repeated inference + shm churn
    with an RSS growth budget."""
    http_host, http_port, _ = http_fixture_server
    grpc_host, grpc_port, _ = grpc_fixture_server
    binary = _compile("memory_leak_test",
                      CPP / "tests" / "memory_leak_test.cc")
    proc = subprocess.run(
        [str(binary), f"{http_host}:{http_port}", f"{grpc_host}:{grpc_port}",
         "300"],
        capture_output=True, text=True, timeout=300,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "ALL PASSED" in proc.stdout
