#!/usr/bin/env python3
"""Health + metadata endpoints (reference: simple_http_health_metadata.py)."""
import argparse

import tritonclient.http as httpclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8000")
    args = parser.parse_args()

    with httpclient.InferenceServerClient(args.url) as client:
        assert client.is_server_live()
        assert client.is_server_ready()
        assert client.is_model_ready("simple")
        print(client.get_server_metadata())
        print(client.get_model_metadata("simple"))
        print(client.get_model_config("simple"))
        print("PASS: health metadata")
