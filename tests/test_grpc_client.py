"""gRPC client <-> server integration tests (CPU only).

Mirrors the reference's typed client test matrix
(cc_client_test.cc:42-129 runs the same suite against HTTP and gRPC)
plus streaming/decoupled coverage.
"""

import queue
import threading

import numpy as np
import pytest

import client_amd.grpc as grpcclient
from client_amd.utils import InferenceServerException


@pytest.fixture()
def client(grpc_fixture_server):
    host, port, _ = grpc_fixture_server
    c = grpcclient.InferenceServerClient(f"{host}:{port}")
    yield c
    c.close()


def test_health(client):
    assert client.is_server_live()
    assert client.is_server_ready()
    assert client.is_model_ready("identity_fp32")
    assert not client.is_model_ready("nope")


def test_metadata_config(client):
    meta = client.get_server_metadata()
    assert meta.name == "client_amd_server"
    as_json = client.get_server_metadata(as_json=True)
    assert as_json["name"] == "client_amd_server"
    mm = client.get_model_metadata("simple")
    assert mm.inputs[0].name == "INPUT0"
    cfg = client.get_model_config("simple")
    assert cfg.config.name == "simple"
    assert cfg.config.input[0].data_type == 8  # TYPE_INT32


def test_repository(client):
    index = client.get_model_repository_index()
    names = {m.name for m in index.models}
    assert "simple" in names
    client.unload_model("simple")
    assert not client.is_model_ready("simple")
    client.load_model("simple")
    assert client.is_model_ready("simple")


def test_infer(client):
    a = np.arange(16, dtype=np.int32).reshape(1, 16)
    b = np.full((1, 16), 3, dtype=np.int32)
    inputs = [
        grpcclient.InferInput("INPUT0", [1, 16], "INT32"),
        grpcclient.InferInput("INPUT1", [1, 16], "INT32"),
    ]
    inputs[0].set_data_from_numpy(a)
    inputs[1].set_data_from_numpy(b)
    outputs = [
        grpcclient.InferRequestedOutput("OUTPUT0"),
        grpcclient.InferRequestedOutput("OUTPUT1"),
    ]
    result = client.infer("simple", inputs, outputs=outputs, request_id="7")
    np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), a + b)
    np.testing.assert_array_equal(result.as_numpy("OUTPUT1"), a - b)
    assert result.get_response().id == "7"
    out_json = result.get_output("OUTPUT0", as_json=True)
    assert out_json["datatype"] == "INT32"


def test_infer_bf16_and_bytes(client):
    x = np.array([[0.5, -1.5, 2.0, 8.0]], dtype=np.float32)
    inp = grpcclient.InferInput("INPUT0", list(x.shape), "BF16")
    inp.set_data_from_numpy(x)
    result = client.infer("identity_bf16", [inp])
    np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x)

    s = np.array([b"ab", b"\xff\x00"], dtype=np.object_)
    inp = grpcclient.InferInput("INPUT0", [2], "BYTES")
    inp.set_data_from_numpy(s)
    result = client.infer("identity_bytes", [inp])
    np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), s)


def test_async_infer(client):
    a = np.arange(16, dtype=np.int32).reshape(1, 16)
    inputs = [
        grpcclient.InferInput("INPUT0", [1, 16], "INT32"),
        grpcclient.InferInput("INPUT1", [1, 16], "INT32"),
    ]
    inputs[0].set_data_from_numpy(a)
    inputs[1].set_data_from_numpy(a)
    done = queue.Queue()

    def callback(result, error):
        done.put((result, error))

    ctx = client.async_infer("simple", inputs, callback)
    assert ctx is not None
    result, error = done.get(timeout=10)
    assert error is None
    np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), a + a)


def test_async_infer_error(client):
    a = np.zeros((1, 16), dtype=np.int32)
    inputs = [grpcclient.InferInput("INPUT0", [1, 16], "INT32")]
    inputs[0].set_data_from_numpy(a)
    done = queue.Queue()
    client.async_infer("no_such_model", inputs, lambda result, error: done.put(error))
    error = done.get(timeout=10)
    assert isinstance(error, InferenceServerException)


def test_stream_sequence(client):
    """Sequence inference over the bi-di stream (reference example:
    simple_grpc_sequence_stream_infer_client)."""
    results = queue.Queue()
    client.start_stream(callback=lambda result, error: results.put((result, error)))
    try:
        for i, (start, end) in enumerate([(True, False), (False, False),
                                          (False, True)]):
            inp = grpcclient.InferInput("INPUT", [1], "INT32")
            inp.set_data_from_numpy(np.array([i + 1], dtype=np.int32))
            client.async_stream_infer(
                "sequence_accumulate", [inp], sequence_id=5,
                sequence_start=start, sequence_end=end,
            )
        vals = []
        for _ in range(3):
            result, error = results.get(timeout=10)
            assert error is None
            vals.append(int(result.as_numpy("OUTPUT")[0]))
        assert vals == [1, 3, 6]
    finally:
        client.stop_stream()


def test_stream_decoupled_repeat(client):
    """Decoupled model: N responses per request, final-response flag
    (reference simple_grpc_custom_repeat.cc:135-176)."""
    results = queue.Queue()
    client.start_stream(callback=lambda result, error: results.put((result, error)))
    try:
        n = 4
        in_vals = np.arange(n, dtype=np.int32)
        delays = np.zeros(n, dtype=np.uint32)
        wait = np.zeros(1, dtype=np.uint32)
        inputs = [
            grpcclient.InferInput("IN", [n], "INT32"),
            grpcclient.InferInput("DELAY", [n], "UINT32"),
            grpcclient.InferInput("WAIT", [1], "UINT32"),
        ]
        inputs[0].set_data_from_numpy(in_vals)
        inputs[1].set_data_from_numpy(delays)
        inputs[2].set_data_from_numpy(wait)
        client.async_stream_infer(
            "repeat_int32", inputs, enable_empty_final_response=True
        )
        seen = []
        while True:
            result, error = results.get(timeout=10)
            assert error is None
            if result.is_final_response():
                break
            seen.append(int(result.as_numpy("OUT")[0]))
        assert seen == list(range(n))
    finally:
        client.stop_stream()


def test_stream_error_in_band(client):
    """A bad request on the stream reports via callback error and the
    stream stays usable."""
    results = queue.Queue()
    client.start_stream(callback=lambda result, error: results.put((result, error)))
    try:
        inp = grpcclient.InferInput("INPUT", [1], "INT32")
        inp.set_data_from_numpy(np.array([1], dtype=np.int32))
        client.async_stream_infer("no_such_model", [inp])
        result, error = results.get(timeout=10)
        assert isinstance(error, InferenceServerException)
        # stream still active
        client.async_stream_infer(
            "sequence_accumulate", [inp], sequence_id=6,
            sequence_start=True, sequence_end=True,
        )
        result, error = results.get(timeout=10)
        assert error is None
    finally:
        client.stop_stream()


def test_statistics(client):
    x = np.random.rand(1, 4).astype(np.float32)
    inp = grpcclient.InferInput("INPUT0", [1, 4], "FP32")
    inp.set_data_from_numpy(x)
    client.infer("identity_fp32", [inp])
    stats = client.get_inference_statistics("identity_fp32")
    assert stats.model_stats[0].inference_count >= 1
    assert stats.model_stats[0].inference_stats.success.count >= 1


def test_trace_log_settings(client):
    settings = client.get_trace_settings()
    assert "trace_rate" in settings.settings
    updated = client.update_trace_settings(settings={"trace_rate": "250"})
    assert updated.settings["trace_rate"].value[0] == "250"
    log = client.get_log_settings(as_json=True)
    assert "log_info" in log["settings"]
    updated = client.update_log_settings({"log_verbose_level": 3})
    assert updated.settings["log_verbose_level"].uint32_param == 3


def test_error_mapping(client):
    with pytest.raises(InferenceServerException) as exc:
        client.get_model_metadata("not_a_model")
    assert "not_a_model" in str(exc.value)


def test_keepalive_and_timeout():
    opts = grpcclient.KeepAliveOptions(keepalive_time_ms=10000)
    c = grpcclient.InferenceServerClient("127.0.0.1:1", keepalive_options=opts)
    with pytest.raises(InferenceServerException):
        c.is_server_live(client_timeout=0.2)
    c.close()


def test_infer_timeout(client, grpc_fixture_server):
    # microscopic timeout -> Deadline Exceeded surface
    # (reference client_timeout_test.cc drives every API this way)
    # drive a model that sleeps server-side (repeat_int32 DELAY) so the
    # deadline reliably expires
    inputs = [
        grpcclient.InferInput("IN", [1], "INT32"),
        grpcclient.InferInput("DELAY", [1], "UINT32"),
        grpcclient.InferInput("WAIT", [1], "UINT32"),
    ]
    inputs[0].set_data_from_numpy(np.array([1], dtype=np.int32))
    inputs[1].set_data_from_numpy(np.array([500], dtype=np.uint32))
    inputs[2].set_data_from_numpy(np.array([0], dtype=np.uint32))
    try:
        client.infer("repeat_int32", inputs, client_timeout=0.05)
        raise AssertionError("expected timeout")
    except InferenceServerException as e:
        assert "DEADLINE" in str(e.status()).upper()


def test_large_tensor_roundtrip(client):
    """16 MB tensor through grpc (message-size limits raised to INT32_MAX
    as in the reference)."""
    x = np.random.rand(4, 1024, 1024).astype(np.float32)
    inp = grpcclient.InferInput("INPUT0", list(x.shape), "FP32")
    inp.set_data_from_numpy(x)
    result = client.infer("identity_fp32", [inp])
    np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x)


def test_vendored_proto_in_sync():
    """proto/grpc_service.proto must match the runtime schema render."""
    from pathlib import Path

    from client_amd.grpc import _proto_gen

    vendored = (Path(__file__).resolve().parent.parent / "proto" /
                "grpc_service.proto").read_text()
    assert vendored == _proto_gen.render(), (
        "regenerate with: python -m client_amd.grpc._proto_gen")


def test_model_config_full_schema_roundtrip():
    """Full reference model_config.proto surface: every top-level section
    of ModelConfig (model_config.proto:1971-2180) is settable and
    survives a serialize/parse roundtrip with correct field numbers."""
    from client_amd.grpc._proto import service_pb2

    cfg = service_pb2.ModelConfig()
    cfg.name = "m"
    cfg.backend = "hip"
    cfg.runtime = "client_amd"
    cfg.version_policy.latest.num_versions = 2
    cfg.max_batch_size = 32
    inp = cfg.input.add()
    inp.name = "IN"
    inp.data_type = 11  # TYPE_FP32
    inp.dims.extend([3, 224, 224])
    inp.is_non_linear_format_io = True
    out = cfg.output.add()
    out.name = "OUT"
    out.label_filename = "labels.txt"  # field 4 (reshape is 5)
    bi = cfg.batch_input.add()
    bi.kind = 3  # BATCH_MAX_ELEMENT_COUNT_AS_SHAPE
    bi.target_name.append("ragged_shape")
    cfg.optimization.cuda.graphs = True
    spec = cfg.optimization.cuda.graph_spec.add()
    spec.batch_size = 8
    spec.input["IN"].dim.extend([3, 224, 224])
    cfg.optimization.priority = 1  # PRIORITY_MAX
    cfg.dynamic_batching.preferred_batch_size.extend([4, 8])
    cfg.dynamic_batching.priority_levels = 3
    cfg.dynamic_batching.default_queue_policy.timeout_action = 1  # DELAY
    cfg.dynamic_batching.priority_queue_policy[1].max_queue_size = 16
    ig = cfg.instance_group.add()
    ig.kind = 1  # KIND_GPU
    ig.gpus.extend([0, 1])
    ig.rate_limiter.resources.add().name = "R1"
    cfg.cc_model_filenames["gfx950"] = "model.so"
    cfg.metric_tags["team"] = "serving"
    cfg.parameters["key"].string_value = "val"
    w = cfg.model_warmup.add()
    w.name = "warm"
    w.inputs["IN"].zero_data = True
    cfg.model_transaction_policy.decoupled = True
    cfg.model_repository_agents.agents.add().name = "agent"
    cfg.response_cache.enable = True
    mc = cfg.model_metrics.metric_control.add()
    mc.metric_identifier.family = "latency"
    mc.histogram_options.buckets.extend([0.1, 1.0])

    data = cfg.SerializeToString()
    cfg2 = service_pb2.ModelConfig()
    cfg2.ParseFromString(data)
    assert cfg2.runtime == "client_amd"
    assert cfg2.version_policy.WhichOneof("policy_choice") == "latest"
    assert cfg2.output[0].label_filename == "labels.txt"
    assert cfg2.batch_input[0].kind == 3
    assert list(cfg2.optimization.cuda.graph_spec[0].input["IN"].dim) == \
        [3, 224, 224]
    assert cfg2.dynamic_batching.priority_queue_policy[1].max_queue_size == 16
    assert cfg2.instance_group[0].rate_limiter.resources[0].name == "R1"
    assert cfg2.cc_model_filenames["gfx950"] == "model.so"
    assert cfg2.model_warmup[0].inputs["IN"].zero_data is True
    assert cfg2.model_metrics.metric_control[0].histogram_options.buckets[1] \
        == 1.0
    assert cfg2.WhichOneof("scheduling_choice") == "dynamic_batching"

    # sequence batching oneof replaces dynamic batching
    cfg2.sequence_batching.oldest.max_candidate_sequences = 4
    st = cfg2.sequence_batching.state.add()
    st.input_name = "S_IN"
    st.initial_state.add().zero_data = True
    assert cfg2.WhichOneof("scheduling_choice") == "sequence_batching"
    cfg3 = service_pb2.ModelConfig()
    cfg3.ParseFromString(cfg2.SerializeToString())
    assert cfg3.sequence_batching.oldest.max_candidate_sequences == 4
    assert cfg3.sequence_batching.state[0].initial_state[0].zero_data is True


def test_async_infer_cancel(grpc_fixture_server):
    """CallContext.cancel() on an in-flight delayed request surfaces a
    CANCELLED error to the callback (reference grpc/_client.py:101-117
    semantics) and leaves the client usable."""
    import queue
    import numpy as np

    import client_amd.grpc as grpcclient

    host, port, _ = grpc_fixture_server
    client = grpcclient.InferenceServerClient(f"{host}:{port}")
    try:
        inputs = [
            grpcclient.InferInput("IN", [1], "INT32"),
            grpcclient.InferInput("DELAY", [1], "UINT32"),
        ]
        inputs[0].set_data_from_numpy(np.array([1], dtype=np.int32))
        inputs[1].set_data_from_numpy(np.array([1500], dtype=np.uint32))
        events = queue.Queue()
        ctx = client.async_infer(
            "repeat_int32", inputs,
            callback=lambda result, error: events.put((result, error)),
        )
        assert ctx.cancel() or True  # cancel may race completion
        result, error = events.get(timeout=30)
        # either cancelled (normal) or, if it raced, a completed result
        if error is not None:
            assert "CANCELLED" in str(error).upper() or "cancel" in str(
                error).lower()
        # client still works
        assert client.is_server_live()
    finally:
        client.close()


def test_stop_stream_cancel_requests(grpc_fixture_server):
    """stop_stream(cancel_requests=True) returns promptly with a slow
    decoupled request still in flight."""
    import queue
    import time
    import numpy as np

    import client_amd.grpc as grpcclient

    host, port, _ = grpc_fixture_server
    client = grpcclient.InferenceServerClient(f"{host}:{port}")
    try:
        events = queue.Queue()
        client.start_stream(
            callback=lambda result, error: events.put((result, error)))
        inputs = [
            grpcclient.InferInput("IN", [4], "INT32"),
            grpcclient.InferInput("DELAY", [4], "UINT32"),
        ]
        inputs[0].set_data_from_numpy(np.arange(4, dtype=np.int32))
        inputs[1].set_data_from_numpy(
            np.full(4, 800, dtype=np.uint32))  # 4 x 800ms of responses
        client.async_stream_infer("repeat_int32", inputs)
        time.sleep(0.2)
        t0 = time.monotonic()
        client.stop_stream(cancel_requests=True)
        assert time.monotonic() - t0 < 3.0, "cancel did not cut the stream"
        assert client.is_server_live()
    finally:
        client.close()


def test_string_sequence_id_and_query_params(grpc_fixture_server,
                                             http_fixture_server):
    """String sequence ids ride the InferParameter oneof (gRPC) and the
    JSON parameters (HTTP); query_params must not break HTTP URLs."""
    import numpy as np

    import client_amd.grpc as grpcclient
    import client_amd.http as httpclient

    ghost, gport, _ = grpc_fixture_server
    gc = grpcclient.InferenceServerClient(f"{ghost}:{gport}")
    try:
        total = 0
        for i, v in enumerate([5, 3]):
            inp = grpcclient.InferInput("INPUT", [1], "INT32")
            inp.set_data_from_numpy(np.array([v], dtype=np.int32))
            result = gc.infer(
                "sequence_accumulate", [inp], sequence_id="stream-A",
                sequence_start=(i == 0), sequence_end=(i == 1),
            )
            total += v
            assert int(result.as_numpy("OUTPUT")[0]) == total
    finally:
        gc.close()

    hhost, hport, _ = http_fixture_server
    hc = httpclient.InferenceServerClient(f"{hhost}:{hport}")
    try:
        inp = httpclient.InferInput("INPUT0", [1, 16], "INT32")
        inp.set_data_from_numpy(np.ones((1, 16), dtype=np.int32))
        result = hc.infer(
            "simple", [inp, _second_ones(httpclient)],
            query_params={"trace": "1", "k": "v"},
        )
        assert result.as_numpy("OUTPUT0").sum() == 32
    finally:
        hc.close()


def _second_ones(mod):
    import numpy as np

    inp = mod.InferInput("INPUT1", [1, 16], "INT32")
    inp.set_data_from_numpy(np.ones((1, 16), dtype=np.int32))
    return inp


def test_grpc_compression(grpc_fixture_server):
    """gRPC channel compression (gzip/deflate enums) roundtrips."""
    import numpy as np

    import client_amd.grpc as grpcclient

    host, port, _ = grpc_fixture_server
    client = grpcclient.InferenceServerClient(f"{host}:{port}")
    try:
        x = np.random.rand(1, 1024).astype(np.float32)
        inp = grpcclient.InferInput("INPUT0", list(x.shape), "FP32")
        inp.set_data_from_numpy(x)
        for algo in ("gzip", "deflate"):
            result = client.infer(
                "identity_fp32", [inp], compression_algorithm=algo
            )
            np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x)
    finally:
        client.close()
