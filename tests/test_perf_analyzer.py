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
