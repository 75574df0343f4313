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


def test_modelconfig_cross_decode():
    """C++ full-tree ModelConfig decoder vs the Python runtime schema
    as oracle: a maximal config touching EVERY message of
    model_config.proto (reference proto/model_config.proto:86-2180) is
    serialized by Python and decoded by the pb_config_dump tool; every
    field must round-trip."""
    import json

    from client_amd.grpc._proto import service_pb2

    try:
        dump_bin = _compile("pb_config_dump",
                            CPP / "tests" / "pb_config_dump.cc")
    except subprocess.CalledProcessError as e:
        pytest.fail(f"C++ compile failed:\n{e.stderr}")

    c = service_pb2.ModelConfig()
    c.name = "maximal"
    c.platform = "client_amd"
    c.backend = "hip"
    c.runtime = "mi355x"
    c.max_batch_size = 64
    c.version_policy.specific.versions.extend([3, 7, 11])

    i0 = c.input.add()
    i0.name = "IN0"
    i0.data_type = 11  # TYPE_FP32
    i0.dims.extend([3, 224, 224])
    i0.format = 2      # FORMAT_NCHW
    i0.reshape.shape.extend([1, 3, 224, 224])
    i0.is_shape_tensor = False
    i0.allow_ragged_batch = True
    i0.optional = True
    i0.is_non_linear_format_io = True

    o0 = c.output.add()
    o0.name = "OUT0"
    o0.data_type = 13  # TYPE_BF16-ish enum slot
    o0.dims.extend([1000])
    o0.label_filename = "labels.txt"
    o0.reshape.shape.extend([10, 100])
    o0.is_shape_tensor = True
    o0.is_non_linear_format_io = True

    g = c.instance_group.add()
    g.name = "ig0"
    g.kind = 1
    g.count = 4
    g.gpus.extend([0, 1, 2, 3])
    g.profile.extend(["p0", "p1"])
    g.passive = True
    g.host_policy = "numa0"
    res = g.rate_limiter.resources.add()
    res.name = "R0"
    setattr(res, "global", True)
    res.count = 5
    g.rate_limiter.priority = 9
    sd = g.secondary_devices.add()
    sd.kind = 0
    sd.device_id = 42

    c.default_model_filename = "model.bin"
    c.cc_model_filenames["linux"] = "model_linux.so"
    c.metric_tags["team"] = "infra"
    c.parameters["key"].string_value = "value"

    opt = c.optimization
    opt.graph.level = 2
    opt.priority = 1
    opt.cuda.graphs = True
    opt.cuda.busy_wait_events = True
    opt.cuda.output_copy_stream = True
    spec = opt.cuda.graph_spec.add()
    spec.batch_size = 8
    spec.input["IN0"].dim.extend([3, 224, 224])
    spec.graph_lower_bound.batch_size = 1
    spec.graph_lower_bound.input["IN0"].dim.extend([3, 64, 64])
    acc = opt.execution_accelerators.gpu_execution_accelerator.add()
    acc.name = "mfma"
    acc.parameters["tile"] = "64"
    acc2 = opt.execution_accelerators.cpu_execution_accelerator.add()
    acc2.name = "cpu0"
    opt.input_pinned_memory.enable = True
    opt.output_pinned_memory.enable = False
    opt.gather_kernel_buffer_threshold = 7
    opt.eager_batching = True

    sb = c.sequence_batching
    sb.max_sequence_idle_microseconds = 1000000
    sb.iterative_sequence = True
    sb.oldest.max_candidate_sequences = 12
    sb.oldest.preferred_batch_size.extend([4, 8])
    sb.oldest.max_queue_delay_microseconds = 500
    sb.oldest.preserve_ordering = True
    ci = sb.control_input.add()
    ci.name = "START"
    ctl = ci.control.add()
    ctl.kind = 0
    ctl.int32_false_true.extend([0, 1])
    ctl.data_type = 8
    ctl2 = ci.control.add()
    ctl2.kind = 2
    ctl2.fp32_false_true.extend([0.0, 1.0])
    ctl3 = ci.control.add()
    ctl3.kind = 1
    ctl3.bool_false_true.extend([False, True])
    st = sb.state.add()
    st.input_name = "S_IN"
    st.output_name = "S_OUT"
    st.data_type = 11
    st.dims.extend([256])
    st.use_same_buffer_for_input_output = True
    st.use_growable_memory = True
    ist = st.initial_state.add()
    ist.name = "init0"
    ist.data_type = 11
    ist.dims.extend([256])
    ist.zero_data = True

    w = c.model_warmup.add()
    w.name = "warm0"
    w.batch_size = 8
    w.count = 3
    wi = w.inputs["IN0"]
    wi.data_type = 11
    wi.dims.extend([3, 224, 224])
    wi.random_data = True

    bi = c.batch_input.add()
    bi.kind = 3
    bi.target_name.append("RAGGED_SHAPE")
    bi.data_type = 8
    bi.source_input.append("IN0")
    bo = c.batch_output.add()
    bo.kind = 0
    bo.target_name.append("OUT0")
    bo.source_input.append("IN0")

    c.model_operations.op_library_filename.append("libops.so")
    c.model_transaction_policy.decoupled = True
    ag = c.model_repository_agents.agents.add()
    ag.name = "checksum"
    ag.parameters["algo"] = "sha256"
    c.response_cache.enable = True
    mc = c.model_metrics.metric_control.add()
    mc.metric_identifier.family = "nv_inference_count"
    mc.histogram_options.buckets.extend([0.5, 1.0, 2.5])

    resp = service_pb2.ModelConfigResponse()
    resp.config.CopyFrom(c)
    wire = resp.SerializeToString()

    proc = subprocess.run([str(dump_bin)], input=wire,
                          capture_output=True, timeout=30)
    assert proc.returncode == 0, proc.stderr.decode()
    d = json.loads(proc.stdout.decode())

    assert d["name"] == "maximal"
    assert d["platform"] == "client_amd"
    assert d["backend"] == "hip"
    assert d["runtime"] == "mi355x"
    assert d["max_batch_size"] == 64
    assert d["version_policy_choice"] == 3  # SPECIFIC
    assert d["version_policy_specific"] == [3, 7, 11]

    di = d["input"][0]
    assert di["name"] == "IN0" and di["data_type"] == 11
    assert di["dims"] == [3, 224, 224] and di["format"] == 2
    assert di["has_reshape"] and di["reshape"] == [1, 3, 224, 224]
    assert di["allow_ragged_batch"] and di["optional"]
    assert di["is_non_linear_format_io"]

    do = d["output"][0]
    assert do["name"] == "OUT0" and do["label_filename"] == "labels.txt"
    assert do["dims"] == [1000] and do["reshape"] == [10, 100]
    assert do["is_shape_tensor"] and do["is_non_linear_format_io"]

    dg = d["instance_group"][0]
    assert dg["name"] == "ig0" and dg["kind"] == 1 and dg["count"] == 4
    assert dg["gpus"] == [0, 1, 2, 3] and dg["profile"] == ["p0", "p1"]
    assert dg["passive"] and dg["host_policy"] == "numa0"
    assert dg["has_rate_limiter"] and dg["rate_limiter_priority"] == 9
    assert dg["rate_limiter_resources"][0]["name"] == "R0"
    assert dg["rate_limiter_resources"][0]["count"] == 5
    assert dg["secondary_devices"][0]["device_id"] == 42

    assert d["default_model_filename"] == "model.bin"
    assert d["cc_model_filenames"] == {"linux": "model_linux.so"}
    assert d["metric_tags"] == {"team": "infra"}
    assert d["parameters"] == {"key": "value"}

    od = d["optimization"]
    assert d["has_optimization"]
    assert od["has_graph"] and od["graph_level"] == 2
    assert od["priority"] == 1
    assert od["has_cuda"] and od["cuda_graphs"]
    assert od["cuda_busy_wait_events"] and od["cuda_output_copy_stream"]
    gs = od["cuda_graph_spec"][0]
    assert gs["batch_size"] == 8 and gs["input"]["IN0"] == [3, 224, 224]
    assert gs["has_lower_bound"] and gs["lower_bound_batch_size"] == 1
    assert od["gpu_execution_accelerator"][0]["name"] == "mfma"
    assert od["gpu_execution_accelerator"][0]["parameters"] == {"tile": "64"}
    assert od["cpu_execution_accelerator"][0]["name"] == "cpu0"
    assert od["has_input_pinned_memory"] and od["input_pinned_memory"]
    assert od["has_output_pinned_memory"] and not od["output_pinned_memory"]
    assert od["gather_kernel_buffer_threshold"] == 7
    assert od["eager_batching"]

    sd2 = d["sequence_batching"]
    assert d["has_sequence_batching"]
    assert sd2["strategy"] == 2  # OLDEST
    assert sd2["oldest_max_candidate_sequences"] == 12
    assert sd2["oldest_preferred_batch_size"] == [4, 8]
    assert sd2["oldest_max_queue_delay_microseconds"] == 500
    assert sd2["oldest_preserve_ordering"]
    assert sd2["max_sequence_idle_microseconds"] == 1000000
    assert sd2["iterative_sequence"]
    dci = sd2["control_input"][0]
    assert dci["name"] == "START"
    assert dci["control"][0]["int32_false_true"] == [0, 1]
    assert dci["control"][0]["data_type"] == 8
    assert dci["control"][1]["fp32_false_true"] == [0.0, 1.0]
    assert dci["control"][2]["bool_false_true"] == [False, True]
    dst = sd2["state"][0]
    assert dst["input_name"] == "S_IN" and dst["output_name"] == "S_OUT"
    assert dst["dims"] == [256]
    assert dst["use_same_buffer_for_input_output"]
    assert dst["use_growable_memory"]
    assert dst["initial_state"][0]["name"] == "init0"
    assert dst["initial_state"][0]["data_choice"] == 1  # ZERO
    assert dst["initial_state"][0]["zero_data"]

    dw = d["model_warmup"][0]
    assert dw["name"] == "warm0" and dw["batch_size"] == 8
    assert dw["count"] == 3
    assert dw["inputs"]["IN0"]["dims"] == [3, 224, 224]
    assert dw["inputs"]["IN0"]["data_choice"] == 2  # RANDOM
    assert dw["inputs"]["IN0"]["random_data"]

    assert d["batch_input"][0]["kind"] == 3
    assert d["batch_input"][0]["target_name"] == ["RAGGED_SHAPE"]
    assert d["batch_input"][0]["source_input"] == ["IN0"]
    assert d["batch_output"][0]["target_name"] == ["OUT0"]

    assert d["op_library_filename"] == ["libops.so"]
    assert d["decoupled"]
    assert d["repository_agents"][0]["name"] == "checksum"
    assert d["repository_agents"][0]["parameters"] == {"algo": "sha256"}
    assert d["response_cache_enable"]
    assert d["metric_control"][0]["family"] == "nv_inference_count"
    assert d["metric_control"][0]["histogram_buckets"] == [0.5, 1.0, 2.5]


@pytest.fixture(scope="module")
def cc_dual_binary():
    try:
        return _compile("cc_dual_test", CPP / "tests" / "cc_dual_test.cc")
    except subprocess.CalledProcessError as e:
        pytest.fail(f"C++ compile failed:\n{e.stderr}")


def test_cc_dual_suite(cc_dual_binary, http_fixture_server,
                       grpc_fixture_server):
    """The reference's typed dual-protocol test matrix (InferMulti /
    AsyncInferMulti option+output combinations, mismatch errors,
    unknown-output and wrong-shape server errors, load with
    config/file override) run against BOTH clients from one
    parameterized C++ suite (reference cc_client_test.cc:42-129,
    300-1350)."""
    hhost, hport, _ = http_fixture_server
    ghost, gport, _ = grpc_fixture_server
    proc = subprocess.run(
        [str(cc_dual_binary), f"{hhost}:{hport}", f"{ghost}:{gport}"],
        capture_output=True, text=True, timeout=180,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "ALL PASSED" in proc.stdout
