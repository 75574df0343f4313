#!/usr/bin/env python3
"""gRPC model control (reference: simple_grpc_model_control.py)."""
import argparse

import tritonclient.grpc as grpcclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()

    with grpcclient.InferenceServerClient(args.url) as client:
        client.unload_model("simple")
        assert not client.is_model_ready("simple")
        client.load_model("simple")
        assert client.is_model_ready("simple")
        print(client.get_model_repository_index())
        print("PASS: grpc model control")
