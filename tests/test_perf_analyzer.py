"""Load-generator tests against the CPU fixture server (wire mode)."""

import numpy as np

from client_amd.perf import PerfAnalyzer


def test_perf_grpc_sweep(grpc_fixture_server):
    host, port, _ = grpc_fixture_server
    pa = PerfAnalyzer(
        url=f"{host}:{port}", protocol="grpc", model_name="simple",
        batch_size=1,
    )
    results = pa.run([1, 2], warmup_s=0.1, window_s=0.25, max_windows=2)
    assert len(results) == 2
    for r in results:
        assert r["errors"] == 0
        assert r["request_rate_per_sec"] > 0
        assert r["latency_us"]["p99"] >= r["latency_us"]["p50"] > 0


def test_perf_http_sweep(http_fixture_server):
    host, port, _ = http_fixture_server
    pa = PerfAnalyzer(
        url=f"{host}:{port}", protocol="http", model_name="identity_fp32",
        batch_size=4,
    )
    results = pa.run([2], warmup_s=0.1, window_s=0.25, max_windows=2)
    r = results[0]
    assert r["errors"] == 0
    assert r["inferences_per_sec"] == r["request_rate_per_sec"] * 4


def test_perf_cli(grpc_fixture_server, tmp_path, capsys):
    import client_amd.perf.__main__ as cli

    host, port, _ = grpc_fixture_server
    out = tmp_path / "results.json"
    cli.main([
        "-m", "simple", "-u", f"{host}:{port}", "-i", "grpc",
        "--concurrency-range", "1:1:1", "--measurement-interval", "0.2",
        "--warmup", "0.1", "--max-windows", "2", "--json", str(out),
        "-f", str(tmp_path / "report.csv"),
    ])
    captured = capsys.readouterr()
    assert "infer/sec" in captured.out
    import json

    data = json.loads(out.read_text())
    assert data[0]["concurrency"] == 1
    csv_text = (tmp_path / "report.csv").read_text()
    assert csv_text.startswith("Concurrency,Inferences/Second")
    assert len(csv_text.strip().splitlines()) == 2


def test_genai_perf_llm_stream():
    """genai-perf-class LLM metrics against the tiny llama served on CPU."""
    from client_amd.perf.genai import GenAiPerf
    from client_amd.server.__main__ import build_core
    from client_amd.server.grpc_server import GrpcServer

    core = build_core(["llama_tiny"], device="cpu", dtype="fp32")
    server = GrpcServer(core, host="127.0.0.1", port=0)
    server.start()
    try:
        ga = GenAiPerf(
            url=f"127.0.0.1:{server.port}", model_name="llama_tiny",
            prompt_tokens=8, output_tokens=4, vocab_size=256,
        )
        result = ga.run(concurrency=2, requests_per_stream=2)
        assert result["errors"] == 0
        assert result["total_output_tokens"] == 2 * 2 * 4
        assert result["ttft_ms"]["p50"] > 0
        assert result["output_tokens_per_sec"] > 0
    finally:
        server.stop(grace=1)


def test_perf_request_rate_mode(grpc_fixture_server):
    """Open-loop rate mode hits the offered rate (fast fixture model) and
    reports rate-mode fields."""
    host, port, _ = grpc_fixture_server
    pa = PerfAnalyzer(
        url=f"{host}:{port}", protocol="grpc", model_name="simple",
        batch_size=1,
    )
    results = pa.run_request_rate(
        [40.0], warmup_s=0.2, window_s=0.5, max_windows=3, max_threads=8
    )
    r = results[0]
    assert r["errors"] == 0
    assert r["target_request_rate"] == 40.0
    assert r["request_distribution"] == "constant"
    # achieved rate within 25% of offered (CPU fixture, loose bound)
    assert 30.0 <= r["request_rate_per_sec"] <= 50.0


def test_perf_request_rate_poisson(grpc_fixture_server):
    host, port, _ = grpc_fixture_server
    pa = PerfAnalyzer(
        url=f"{host}:{port}", protocol="grpc", model_name="simple",
        batch_size=1, percentile_q=95,
    )
    results = pa.run_request_rate(
        [30.0], warmup_s=0.2, window_s=0.5, max_windows=2,
        distribution="poisson", max_threads=8,
    )
    r = results[0]
    assert r["errors"] == 0
    assert "p95" in r["latency_us"]  # --percentile extra report
    assert 15.0 <= r["request_rate_per_sec"] <= 45.0


def test_perf_input_data_file(grpc_fixture_server, tmp_path):
    """--input-data JSON: provided tensors are used instead of synthetic
    (verified via the addsub fixture semantics on a known vector)."""
    import json

    from client_amd.perf.analyzer import load_input_data

    data_file = tmp_path / "inputs.json"
    data_file.write_text(json.dumps({
        "data": [
            {"INPUT0": {"content": list(range(16)), "shape": [1, 16]},
             "INPUT1": list(range(16))},
        ]
    }))
    entries = load_input_data(str(data_file))
    assert entries[0]["INPUT0"][1] == [1, 16]
    assert entries[0]["INPUT1"][0] == list(range(16))

    host, port, _ = grpc_fixture_server
    pa = PerfAnalyzer(
        url=f"{host}:{port}", protocol="grpc", model_name="simple",
        batch_size=1, input_data=str(data_file),
    )
    # slot arrays must carry the file contents
    client, mod = pa._make_client(1)
    try:
        inputs, outputs = pa._model_io(client)
        slots = pa._setup_wire_slots(mod, inputs, outputs, 1)
        arr = slots[0][0][0]._raw_content  # INPUT0 wire bytes
        np.testing.assert_array_equal(
            np.frombuffer(arr, dtype=np.int32), np.arange(16, dtype=np.int32)
        )
    finally:
        client.close()
    results = pa.run([1], warmup_s=0.1, window_s=0.25, max_windows=2)
    assert results[0]["errors"] == 0


def test_perf_cli_request_rate(grpc_fixture_server, tmp_path, capsys):
    import client_amd.perf.__main__ as cli

    host, port, _ = grpc_fixture_server
    cli.main([
        "-m", "simple", "-u", f"{host}:{port}",
        "--request-rate-range", "20:20:1", "--request-distribution",
        "poisson", "--measurement-interval", "0.3", "--warmup", "0.1",
        "--max-windows", "2", "-f", str(tmp_path / "rate.csv"),
    ])
    captured = capsys.readouterr()
    assert "Request rate: 20" in captured.out
    assert (tmp_path / "rate.csv").read_text().startswith(
        "Request Rate,Inferences/Second")


def test_perf_server_breakdown(grpc_fixture_server):
    """Result includes perf_analyzer's server-side queue/compute
    breakdown derived from KServe-v2 statistics deltas."""
    host, port, _ = grpc_fixture_server
    pa = PerfAnalyzer(
        url=f"{host}:{port}", protocol="grpc", model_name="simple",
        batch_size=1,
    )
    r = pa.run([2], warmup_s=0.1, window_s=0.3, max_windows=2)[0]
    assert "server" in r
    srv = r["server"]
    assert srv["requests"] > 0
    for k in ("avg_queue_us", "avg_compute_input_us",
              "avg_compute_infer_us", "avg_compute_output_us"):
        assert srv[k] >= 0


def test_perf_periodic_ramp(grpc_fixture_server):
    """--periodic-concurrency-range analog: one run, concurrency ramps
    1 -> 3 by 1 per period, one result stage per level."""
    host, port, _ = grpc_fixture_server
    pa = PerfAnalyzer(
        url=f"{host}:{port}", protocol="grpc", model_name="simple",
        batch_size=1,
    )
    results = pa.run_periodic(1, 3, 1, period_s=0.3, warmup_s=0.1)
    assert [r["concurrency"] for r in results] == [1, 2, 3]
    for r in results:
        assert r["errors"] == 0
        assert r["ramped"] is True
        assert r["request_rate_per_sec"] > 0


def test_perf_cli_periodic(grpc_fixture_server, tmp_path, capsys):
    import json as _json

    import client_amd.perf.__main__ as cli

    host, port, _ = grpc_fixture_server
    out = tmp_path / "ramp.json"
    csv = tmp_path / "ramp.csv"
    cli.main([
        "-m", "simple", "-u", f"{host}:{port}", "-i", "grpc",
        "--periodic-concurrency-range", "1:2:1",
        "--request-period", "0.3", "--warmup", "0.1",
        "--json", str(out), "-f", str(csv),
    ])
    data = _json.loads(out.read_text())
    assert [r["concurrency"] for r in data] == [1, 2]
    lines = csv.read_text().strip().splitlines()
    assert lines[0].startswith("Concurrency,")
    assert len(lines) == 3
    captured = capsys.readouterr()
    assert "Ramp concurrency: 1" in captured.out


def test_perf_wildcard_dim_warning(grpc_fixture_server, capsys):
    """Defaulted NON-batch wildcard dims must be announced (a silent
    seq16 once masqueraded as a seq128 BERT measurement)."""
    from client_amd.server.models import IdentityModel

    host, port, core = grpc_fixture_server
    wild = IdentityModel(name="identity_wild")
    wild.inputs = [("INPUT0", "FP32", [-1, -1])]
    wild.outputs = [("OUTPUT0", "FP32", [-1, -1])]
    core.add_model(wild)
    pa = PerfAnalyzer(
        url=f"{host}:{port}", protocol="grpc", model_name="identity_wild",
        batch_size=2,
    )
    pa.run([1], warmup_s=0.05, window_s=0.1, max_windows=1)
    err = capsys.readouterr().err
    assert "wildcard dims" in err and "--shape" in err
