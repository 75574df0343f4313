#!/usr/bin/env python3
"""Raw grpc channel_args passthrough (reference: simple_grpc_custom_args_client.py)."""
import argparse

import tritonclient.grpc as grpcclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()

    channel_args = [("grpc.primary_user_agent", "client_amd-example")]
    with grpcclient.InferenceServerClient(args.url, channel_args=channel_args) as c:
        assert c.is_server_live()
        print("PASS: custom args")
