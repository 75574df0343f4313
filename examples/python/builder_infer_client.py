#!/usr/bin/env python3
"""Fluent request-builder inference (the Rust client's
InferRequestBuilder surface — reference
src/rust/triton-client/src/infer.rs:548 — available to both
protocols from one builder)."""
import argparse
import sys

import numpy as np

from client_amd import InferRequestBuilder
import client_amd.grpc as grpcclient
import client_amd.http as httpclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8000")
    parser.add_argument("-i", "--protocol", default="http",
                        choices=("http", "grpc"))
    args = parser.parse_args()

    cls = (httpclient if args.protocol == "http"
           else grpcclient).InferenceServerClient
    x0 = np.arange(16, dtype=np.int32).reshape(1, 16)
    x1 = np.ones((1, 16), dtype=np.int32)
    with cls(args.url) as client:
        result = (InferRequestBuilder("simple")
                  .request_id("builder-demo")
                  .input_from_numpy("INPUT0", x0)
                  .input_from_numpy("INPUT1", x1)
                  .output("OUTPUT0")
                  .output("OUTPUT1")
                  .infer(client))
        if not np.array_equal(result.as_numpy("OUTPUT0"), x0 + x1):
            sys.exit("addsub mismatch")
        if not np.array_equal(result.as_numpy("OUTPUT1"), x0 - x1):
            sys.exit("addsub mismatch")
    print("PASS: builder_infer_client")
