#!/usr/bin/env python3
"""Model load/unload + repository index (reference: simple_http_model_control.py)."""
import argparse

import tritonclient.http as httpclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8000")
    args = parser.parse_args()

    with httpclient.InferenceServerClient(args.url) as client:
        client.unload_model("simple")
        assert not client.is_model_ready("simple")
        client.load_model("simple")
        assert client.is_model_ready("simple")
        print(client.get_model_repository_index())
        print("PASS: model control")
