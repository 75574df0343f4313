#!/usr/bin/env python3
"""Custom gRPC keepalive options (reference: simple_grpc_keepalive_client.py)."""
import argparse

import tritonclient.grpc as grpcclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()

    options = grpcclient.KeepAliveOptions(
        keepalive_time_ms=10000,
        keepalive_timeout_ms=5000,
        keepalive_permit_without_calls=True,
        http2_max_pings_without_data=3,
    )
    with grpcclient.InferenceServerClient(args.url, keepalive_options=options) as c:
        assert c.is_server_live()
        print("PASS: keepalive")
