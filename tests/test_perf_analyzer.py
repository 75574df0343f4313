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
    ])
    captured = capsys.readouterr()
    assert "infer/sec" in captured.out
    import json

    data = json.loads(out.read_text())
    assert data[0]["concurrency"] == 1
