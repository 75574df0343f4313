"""CLI mirroring perf_analyzer's common flags:

    python -m client_amd.perf -m resnet50 -u 127.0.0.1:8001 -i grpc \
        -b 8 --shared-memory cuda --concurrency-range 1:8:1
"""

import argparse
import json


def main(argv=None):
    p = argparse.ArgumentParser("client_amd.perf")
    p.add_argument("-m", "--model-name", required=True)
    p.add_argument("-u", "--url", default="127.0.0.1:8001")
    p.add_argument("-i", "--protocol", default="grpc",
                   choices=["grpc", "http"])
    p.add_argument("-b", "--batch-size", type=int, default=1)
    p.add_argument("--concurrency-range", default=None,
                   help="start:end:step (closed loop; default 1:4:1)")
    p.add_argument("--periodic-concurrency-range", default=None,
                   help="start:end:step — ONE run whose concurrency "
                        "ramps by step every --request-period seconds "
                        "(per-stage stats, connections kept up)")
    p.add_argument("--request-period", type=float, default=2.0,
                   help="seconds per stage of the periodic ramp")
    p.add_argument("--request-rate-range", default=None,
                   help="start:end:step req/s (open loop; overrides "
                        "--concurrency-range)")
    p.add_argument("--request-distribution", default="constant",
                   choices=["constant", "poisson"])
    p.add_argument("--max-threads", type=int, default=16,
                   help="worker threads for request-rate mode")
    p.add_argument("--input-data", default=None,
                   help="JSON file with real request tensors "
                        '({"data": [{name: [...] | {"content","shape"}}]})')
    p.add_argument("--percentile", type=float, default=None,
                   help="additionally report this latency percentile")
    p.add_argument("--shared-memory", default="none",
                   choices=["none", "cuda", "hip"])
    p.add_argument("--repack", action="store_true",
                   help="re-run the pack kernel before every request")
    p.add_argument("--measurement-interval", type=float, default=2.0,
                   help="seconds per measurement window")
    p.add_argument("--warmup", type=float, default=1.0)
    p.add_argument("--max-windows", type=int, default=6)
    p.add_argument("--stability-percentage", type=float, default=10.0)
    p.add_argument("--shape", action="append", default=[],
                   help="NAME:d1,d2 override for dynamic input dims")
    p.add_argument("--json", default=None, help="write results to file")
    p.add_argument("-f", "--csv", default=None,
                   help="write a perf_analyzer-style CSV report")
    p.add_argument("-v", "--verbose", action="store_true")
    args = p.parse_args(argv)

    def parse_range(spec, cast=int):
        parts = spec.split(":")
        start = cast(parts[0])
        end = cast(parts[1]) if len(parts) > 1 else start
        step = cast(parts[2]) if len(parts) > 2 else cast(1)
        out, v = [], start
        while v <= end:
            out.append(v)
            v += step
        return out

    rate_list = (parse_range(args.request_rate_range, float)
                 if args.request_rate_range else None)
    concurrency_list = parse_range(args.concurrency_range or "1:4:1")

    from .analyzer import PerfAnalyzer

    shapes = {}
    for spec in args.shape:
        name, _, dims = spec.partition(":")
        shapes[name] = [int(d) for d in dims.split(",") if d]
    pa = PerfAnalyzer(
        url=args.url,
        protocol=args.protocol,
        model_name=args.model_name,
        batch_size=args.batch_size,
        shared_memory=args.shared_memory,
        repack=args.repack,
        verbose=args.verbose,
        shapes=shapes,
        input_data=args.input_data,
        percentile_q=args.percentile,
    )
    if args.periodic_concurrency_range is not None:
        lo, _, rest = args.periodic_concurrency_range.partition(":")
        hi, _, st = rest.partition(":")
        results = pa.run_periodic(
            int(lo), int(hi or lo), int(st or 1),
            period_s=args.request_period, warmup_s=args.warmup,
        )
        for r in results:
            print(
                f"Ramp concurrency: {r['concurrency']}, throughput: "
                f"{r['inferences_per_sec']} infer/sec, latency p99: "
                f"{r['latency_us']['p99']} usec"
            )
    elif rate_list is not None:
        results = pa.run_request_rate(
            rate_list,
            warmup_s=args.warmup,
            window_s=args.measurement_interval,
            max_windows=args.max_windows,
            stability_pct=args.stability_percentage,
            distribution=args.request_distribution,
            max_threads=args.max_threads,
        )
        for r in results:
            print(
                f"Request rate: {r['target_request_rate']}, achieved: "
                f"{r['request_rate_per_sec']} req/sec, throughput: "
                f"{r['inferences_per_sec']} infer/sec, latency p99: "
                f"{r['latency_us']['p99']} usec"
            )
    else:
        results = pa.run(
            concurrency_list,
            warmup_s=args.warmup,
            window_s=args.measurement_interval,
            max_windows=args.max_windows,
            stability_pct=args.stability_percentage,
        )
        for r in results:
            print(
                f"Concurrency: {r['concurrency']}, throughput: "
                f"{r['inferences_per_sec']} infer/sec, latency p99: "
                f"{r['latency_us']['p99']} usec"
            )
    if args.json:
        with open(args.json, "w") as f:
            json.dump(results, f, indent=2)
    if args.csv:
        with open(args.csv, "w") as f:
            key = ("Concurrency" if rate_list is None
                   else "Request Rate")
            f.write(f"{key},Inferences/Second,Client Send,"
                    "p50 latency,p90 latency,p95 latency,p99 latency\n")
            for r in results:
                lat = r["latency_us"]
                first = (r["concurrency"] if rate_list is None
                         else r["target_request_rate"])
                f.write(f"{first},{r['inferences_per_sec']},0,"
                        f"{lat['p50']},{lat['p90']},{lat['p95']},"
                        f"{lat['p99']}\n")


if __name__ == "__main__":
    main()
