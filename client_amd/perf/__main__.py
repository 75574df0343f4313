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
    p.add_argument("--concurrency-range", default="1:4:1",
                   help="start:end:step")
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

    parts = args.concurrency_range.split(":")
    start = int(parts[0])
    end = int(parts[1]) if len(parts) > 1 else start
    step = int(parts[2]) if len(parts) > 2 else 1
    concurrency_list = list(range(start, end + 1, step))

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
    )
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
            f.write("Concurrency,Inferences/Second,Client Send,"
                    "p50 latency,p90 latency,p95 latency,p99 latency\n")
            for r in results:
                lat = r["latency_us"]
                f.write(f"{r['concurrency']},{r['inferences_per_sec']},0,"
                        f"{lat['p50']},{lat['p90']},{lat['p95']},"
                        f"{lat['p99']}\n")


if __name__ == "__main__":
    main()
