#!/usr/bin/env python3
"""gRPC health/metadata (reference: simple_grpc_health_metadata.py)."""
import argparse

import tritonclient.grpc as grpcclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()

    with grpcclient.InferenceServerClient(args.url) as client:
        assert client.is_server_live()
        assert client.is_server_ready()
        assert client.is_model_ready("simple")
        print(client.get_server_metadata())
        print(client.get_model_metadata("simple"))
        print("PASS: grpc health metadata")
