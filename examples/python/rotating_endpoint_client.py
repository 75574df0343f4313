#!/usr/bin/env python3
"""Rotating endpoints + retry (the Java client's endpoint package —
reference src/java/.../endpoint/AbstractEndpoint.java and
InferenceServerClient.java retry loop): requests rotate over a URL
list; transport failures fail over to the next endpoint."""
import argparse
import sys

import numpy as np

from client_amd import InferRequestBuilder, MultiEndpointClient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8000",
                        help="comma-separated list; dead entries are "
                             "failed over")
    args = parser.parse_args()

    urls = args.url.split(",")
    # prepend a dead endpoint to demonstrate failover
    client = MultiEndpointClient(["127.0.0.1:1"] + urls, protocol="http",
                                 retries=len(urls) + 1,
                                 network_timeout=10.0,
                                 connection_timeout=5.0)
    try:
        if not client.is_server_live():
            sys.exit("server not live")
        x0 = np.arange(16, dtype=np.int32).reshape(1, 16)
        x1 = np.ones((1, 16), dtype=np.int32)
        for _ in range(3):  # rotation covers the dead endpoint too
            result = (InferRequestBuilder("simple")
                      .input_from_numpy("INPUT0", x0)
                      .input_from_numpy("INPUT1", x1)
                      .infer(client))
            if not np.array_equal(result.as_numpy("OUTPUT0"), x0 + x1):
                sys.exit("addsub mismatch")
    finally:
        client.close()
    print("PASS: rotating_endpoint_client")
