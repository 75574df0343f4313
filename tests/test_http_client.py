"""HTTP client <-> server integration tests (CPU only).

Covers the tier-2 strategy of SURVEY.md §4: a live fixture server and
the full client surface — health/metadata/config/repo/infer (binary,
JSON, BYTES, BF16), async_infer, compression, statistics, trace/log
settings, error mapping, sequence state, and system-shm registration
plumbing.
"""

import numpy as np
import pytest

import client_amd.http as httpclient
from client_amd.utils import InferenceServerException


@pytest.fixture()
def client(http_fixture_server):
    host, port, _ = http_fixture_server
    c = httpclient.InferenceServerClient(f"{host}:{port}", concurrency=4)
    yield c
    c.close()


def test_health(client):
    assert client.is_server_live()
    assert client.is_server_ready()
    assert client.is_model_ready("identity_fp32")
    assert not client.is_model_ready("nonexistent_model")


def test_server_metadata(client):
    meta = client.get_server_metadata()
    assert meta["name"] == "client_amd_server"
    assert "binary_tensor_data" in meta["extensions"]


def test_model_metadata_and_config(client):
    meta = client.get_model_metadata("simple")
    assert meta["name"] == "simple"
    assert {i["name"] for i in meta["inputs"]} == {"INPUT0", "INPUT1"}
    config = client.get_model_config("simple")
    assert config["name"] == "simple"
    assert config["max_batch_size"] == 0


def test_repository_index_load_unload(client):
    index = client.get_model_repository_index()
    names = {m["name"] for m in index}
    assert "simple" in names
    client.unload_model("simple")
    assert not client.is_model_ready("simple")
    client.load_model("simple")
    assert client.is_model_ready("simple")


def test_infer_binary_addsub(client):
    a = np.arange(16, dtype=np.int32).reshape(1, 16)
    b = np.ones((1, 16), dtype=np.int32)
    inputs = [
        httpclient.InferInput("INPUT0", [1, 16], "INT32"),
        httpclient.InferInput("INPUT1", [1, 16], "INT32"),
    ]
    inputs[0].set_data_from_numpy(a)
    inputs[1].set_data_from_numpy(b)
    outputs = [
        httpclient.InferRequestedOutput("OUTPUT0"),
        httpclient.InferRequestedOutput("OUTPUT1", binary_data=False),
    ]
    result = client.infer("simple", inputs, outputs=outputs, request_id="42")
    np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), a + b)
    np.testing.assert_array_equal(result.as_numpy("OUTPUT1"), a - b)
    assert result.get_response()["id"] == "42"
    assert result.get_output("OUTPUT0")["datatype"] == "INT32"
    assert result.get_output("NOPE") is None


def test_infer_json_path(client):
    a = np.arange(16, dtype=np.int32).reshape(1, 16)
    b = np.full((1, 16), 2, dtype=np.int32)
    inputs = [
        httpclient.InferInput("INPUT0", [1, 16], "INT32"),
        httpclient.InferInput("INPUT1", [1, 16], "INT32"),
    ]
    inputs[0].set_data_from_numpy(a, binary_data=False)
    inputs[1].set_data_from_numpy(b, binary_data=False)
    result = client.infer("simple", inputs)
    np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), a + b)


def test_infer_identity_fp32_no_outputs(client):
    x = np.random.rand(1, 37).astype(np.float32)
    inp = httpclient.InferInput("INPUT0", list(x.shape), "FP32")
    inp.set_data_from_numpy(x)
    result = client.infer("identity_fp32", [inp])
    np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x)


def test_infer_bf16(client):
    x = np.array([[1.0, -2.5, 0.5, 3.0]], dtype=np.float32)
    inp = httpclient.InferInput("INPUT0", list(x.shape), "BF16")
    inp.set_data_from_numpy(x)
    out = httpclient.InferRequestedOutput("OUTPUT0", binary_data=True)
    result = client.infer("identity_bf16", [inp], outputs=[out])
    np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x)


def test_bf16_json_rejected(client):
    x = np.zeros((1, 4), dtype=np.float32)
    inp = httpclient.InferInput("INPUT0", [1, 4], "BF16")
    with pytest.raises(InferenceServerException):
        inp.set_data_from_numpy(x, binary_data=False)


def test_infer_bytes(client):
    x = np.array([b"hello", b"world \xff"], dtype=np.object_)
    inp = httpclient.InferInput("INPUT0", [2], "BYTES")
    inp.set_data_from_numpy(x)
    result = client.infer("identity_bytes", [inp])
    np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x)


def test_infer_string_json(client):
    a = np.array([[str(i) for i in range(16)]], dtype=np.object_)
    b = np.array([["1"] * 16], dtype=np.object_)
    inputs = [
        httpclient.InferInput("INPUT0", [1, 16], "BYTES"),
        httpclient.InferInput("INPUT1", [1, 16], "BYTES"),
    ]
    inputs[0].set_data_from_numpy(a, binary_data=False)
    inputs[1].set_data_from_numpy(b, binary_data=True)
    result = client.infer("simple_string", inputs)
    out0 = result.as_numpy("OUTPUT0")
    assert out0[0, 2] == b"3"


def test_async_infer(client):
    reqs = []
    for i in range(8):
        x = np.full((1, 16), i, dtype=np.int32)
        inputs = [
            httpclient.InferInput("INPUT0", [1, 16], "INT32"),
            httpclient.InferInput("INPUT1", [1, 16], "INT32"),
        ]
        inputs[0].set_data_from_numpy(x)
        inputs[1].set_data_from_numpy(x)
        reqs.append((i, client.async_infer("simple", inputs)))
    for i, req in reqs:
        result = req.get_result()
        np.testing.assert_array_equal(
            result.as_numpy("OUTPUT0"), np.full((1, 16), 2 * i, dtype=np.int32)
        )


def test_compression(client):
    x = np.random.rand(1, 1024).astype(np.float32)
    inp = httpclient.InferInput("INPUT0", list(x.shape), "FP32")
    inp.set_data_from_numpy(x)
    for algo in ("gzip", "deflate"):
        result = client.infer(
            "identity_fp32", [inp], request_compression_algorithm=algo
        )
        np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x)


def test_response_compression(client):
    """response_compression_algorithm: server gzips/deflates the whole
    body (Accept-Encoding), InferResult transparently decompresses."""
    x = np.random.rand(1, 1024).astype(np.float32)
    inp = httpclient.InferInput("INPUT0", list(x.shape), "FP32")
    inp.set_data_from_numpy(x)
    for algo in ("gzip", "deflate"):
        result = client.infer(
            "identity_fp32", [inp], response_compression_algorithm=algo
        )
        assert result.get_response_header("content-encoding") == algo
        np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x)


def test_error_mapping(client):
    with pytest.raises(InferenceServerException) as exc:
        client.get_model_metadata("not_a_model")
    assert "not_a_model" in str(exc.value)

    x = np.zeros((1, 16), dtype=np.int32)
    inputs = [httpclient.InferInput("INPUT0", [1, 16], "INT32")]
    inputs[0].set_data_from_numpy(x)
    with pytest.raises(InferenceServerException):
        client.infer("nonexistent_model", inputs)


def test_statistics(client):
    x = np.random.rand(1, 8).astype(np.float32)
    inp = httpclient.InferInput("INPUT0", [1, 8], "FP32")
    inp.set_data_from_numpy(x)
    client.infer("identity_fp32", [inp])
    stats = client.get_inference_statistics("identity_fp32")
    ms = stats["model_stats"][0]
    assert ms["name"] == "identity_fp32"
    assert ms["inference_count"] >= 1
    assert ms["inference_stats"]["success"]["count"] >= 1


def test_trace_and_log_settings(client):
    settings = client.get_trace_settings()
    assert "trace_rate" in settings
    updated = client.update_trace_settings(settings={"trace_rate": "500"})
    assert updated["trace_rate"] == "500"
    log = client.get_log_settings()
    assert "log_info" in log
    updated = client.update_log_settings({"log_verbose_level": 2})
    assert updated["log_verbose_level"] == 2


def test_sequence(client):
    results = []
    for i, (start, end) in enumerate([(True, False), (False, False), (False, True)]):
        inp = httpclient.InferInput("INPUT", [1], "INT32")
        inp.set_data_from_numpy(np.array([i + 1], dtype=np.int32))
        r = client.infer(
            "sequence_accumulate",
            [inp],
            sequence_id=99,
            sequence_start=start,
            sequence_end=end,
        )
        results.append(int(r.as_numpy("OUTPUT")[0]))
    assert results == [1, 3, 6]


def test_custom_parameters_reserved_rejected(client):
    x = np.zeros((1, 8), dtype=np.float32)
    inp = httpclient.InferInput("INPUT0", [1, 8], "FP32")
    inp.set_data_from_numpy(x)
    with pytest.raises(InferenceServerException):
        client.infer("identity_fp32", [inp], parameters={"sequence_id": 7})


def test_generate_and_parse_body_stateless():
    x = np.arange(4, dtype=np.float32).reshape(1, 4)
    inp = httpclient.InferInput("INPUT0", [1, 4], "FP32")
    inp.set_data_from_numpy(x)
    body, json_size = httpclient.InferenceServerClient.generate_request_body([inp])
    assert json_size is not None
    assert body[json_size:] == x.tobytes()


def test_basic_auth_plugin(client, http_fixture_server):
    from client_amd._auth import BasicAuth

    client.register_plugin(BasicAuth("user", "pass"))
    assert client.plugin() is not None
    # plugin is applied to every request; server ignores the header
    assert client.is_server_live()
    client.unregister_plugin()
    assert client.plugin() is None


def test_transfer_encoding_header_rejected(client):
    with pytest.raises(InferenceServerException):
        client.is_server_live(headers={"Transfer-Encoding": "chunked"})


def test_large_tensor_roundtrip(client):
    """16 MB tensor through the pooled HTTP transport (framing +
    keep-alive reuse at the reference's 16 MiB buffer scale)."""
    x = np.random.rand(4, 1024, 1024).astype(np.float32)  # 16 MiB
    inp = httpclient.InferInput("INPUT0", list(x.shape), "FP32")
    inp.set_data_from_numpy(x)
    result = client.infer("identity_fp32", [inp])
    np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x)
    # connection reuse after the big transfer
    assert client.is_server_live()


def test_orca_load_metrics_header():
    """Server emits ORCA endpoint-load-metrics when the client opts in
    via endpoint-load-metrics-format (reference README.md:352-366);
    surfaced by InferResult.get_response_header()."""
    import numpy as np

    import client_amd.http as httpclient
    from client_amd.server import HttpServer
    from client_amd.server.__main__ import build_core

    core = build_core(["llama_tiny"], device="cpu", dtype="fp32")
    server = HttpServer(core, host="127.0.0.1", port=0)
    stop = server.serve_forever_in_thread()
    try:
        client = httpclient.InferenceServerClient(f"127.0.0.1:{server.port}")
        inputs = [
            httpclient.InferInput("input_ids", [4], "INT64"),
            httpclient.InferInput("max_tokens", [1], "INT32"),
        ]
        inputs[0].set_data_from_numpy(np.array([1, 2, 3, 4], dtype=np.int64))
        inputs[1].set_data_from_numpy(np.array([2], dtype=np.int32))
        result = client.infer(
            "llama_tiny", inputs,
            headers={"endpoint-load-metrics-format": "text"},
        )
        hdr = result.get_response_header("endpoint-load-metrics")
        assert hdr is not None and hdr.startswith("TEXT ")
        assert "kv_cache_utilization=" in hdr
        result_json = client.infer(
            "llama_tiny", inputs,
            headers={"endpoint-load-metrics-format": "json"},
        )
        hdr2 = result_json.get_response_header("endpoint-load-metrics")
        assert hdr2.startswith("JSON ")
        # no opt-in -> no header
        result_none = client.infer("llama_tiny", inputs)
        assert result_none.get_response_header("endpoint-load-metrics") is None
        client.close()
    finally:
        stop()


def test_network_timeout(http_fixture_server):
    """Client-side network_timeout on a response the server delays past
    it maps to an InferenceServerException (timeout taxonomy)."""
    import numpy as np

    host, port, _ = http_fixture_server
    slow = httpclient.InferenceServerClient(
        f"{host}:{port}", network_timeout=0.3)
    try:
        inputs = [
            httpclient.InferInput("IN", [1], "INT32"),
            httpclient.InferInput("DELAY", [1], "UINT32"),
        ]
        inputs[0].set_data_from_numpy(np.array([1], dtype=np.int32))
        inputs[1].set_data_from_numpy(np.array([2000], dtype=np.uint32))
        with pytest.raises(InferenceServerException):
            slow.infer("repeat_int32", inputs)
    finally:
        slow.close()


def test_parse_response_body_roundtrip(client, http_fixture_server):
    """generate_request_body -> raw POST -> parse_response_body: the
    stateless out-of-band pair (the perf_analyzer hook surface)."""
    import http.client as stdhttp

    import numpy as np

    host, port, _ = http_fixture_server
    x = np.arange(16, dtype=np.float32).reshape(1, 16)
    inp = httpclient.InferInput("INPUT0", [1, 16], "FP32")
    inp.set_data_from_numpy(x)
    body, json_size = httpclient.InferenceServerClient.generate_request_body(
        [inp])

    conn = stdhttp.HTTPConnection(host, port, timeout=30)
    try:
        conn.request(
            "POST", "/v2/models/identity_fp32/infer", body=body,
            headers={"Inference-Header-Content-Length": str(json_size),
                     "Content-Type": "application/octet-stream"},
        )
        resp = conn.getresponse()
        assert resp.status == 200
        header_len = resp.headers.get("Inference-Header-Content-Length")
        raw = resp.read()
    finally:
        conn.close()

    result = httpclient.InferenceServerClient.parse_response_body(
        raw, header_length=int(header_len) if header_len else None)
    np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x)


def test_bf16_json_rejected_fp16_allowed():
    """Only BF16 is binary-only over HTTP (it has no JSON number form);
    FP16 inlines into JSON like any float — the reference restricts
    exactly BF16 (reference http/_infer_input.py:169-171)."""
    import numpy as np

    inp = httpclient.InferInput("X", [2], "BF16")
    with pytest.raises(InferenceServerException):
        inp.set_data_from_numpy(np.zeros(2, np.float32), binary_data=False)
    inp16 = httpclient.InferInput("X", [2], "FP16")
    inp16.set_data_from_numpy(
        np.array([1.5, -2.0], np.float16), binary_data=False
    )
    assert inp16._get_tensor()["data"] == [1.5, -2.0]
    assert inp16._get_binary_data() is None
