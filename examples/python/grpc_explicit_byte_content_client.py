#!/usr/bin/env python3
"""Explicit BYTES tensor contents via InferTensorContents.bytes_contents
(one bytes value per element — no length-prefixed packing needed)
(reference: grpc_explicit_byte_content_client.py)."""
import argparse

import grpc
import numpy as np

from client_amd.grpc._proto import RPCS, SERVICE_NAME, service_pb2
from tritonclient.utils import deserialize_bytes_tensor

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()

    channel = grpc.insecure_channel(args.url)
    req_cls, resp_cls, _ = RPCS["ModelInfer"]
    infer = channel.unary_unary(
        f"/{SERVICE_NAME}/ModelInfer",
        request_serializer=req_cls.SerializeToString,
        response_deserializer=resp_cls.FromString,
    )
    request = service_pb2.ModelInferRequest()
    request.model_name = "simple_string"
    for name, vals in (
        ("INPUT0", [str(i).encode() for i in range(16)]),
        ("INPUT1", [b"1"] * 16),
    ):
        t = request.inputs.add()
        t.name = name
        t.datatype = "BYTES"
        t.shape.extend([1, 16])
        t.contents.bytes_contents.extend(vals)
    response = infer(request)
    out0 = deserialize_bytes_tensor(response.raw_output_contents[0])
    got = np.array([int(v) for v in out0.reshape(-1)])
    assert (got == np.arange(16) + 1).all()
    channel.close()
    print("PASS: explicit bytes contents")
