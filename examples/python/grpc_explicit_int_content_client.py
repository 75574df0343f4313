#!/usr/bin/env python3
"""Explicit typed tensor contents: INT32 data in
InferTensorContents.int_contents instead of raw bytes
(reference: grpc_explicit_int_content_client.py)."""
import argparse

import grpc
import numpy as np

from client_amd.grpc._proto import RPCS, SERVICE_NAME, service_pb2

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
    request.model_name = "simple"
    in0 = list(range(16))
    in1 = [1] * 16
    for name, vals in (("INPUT0", in0), ("INPUT1", in1)):
        t = request.inputs.add()
        t.name = name
        t.datatype = "INT32"
        t.shape.extend([1, 16])
        t.contents.int_contents.extend(vals)
    response = infer(request)
    out0 = np.frombuffer(response.raw_output_contents[0], dtype=np.int32)
    assert (out0 == np.array(in0) + np.array(in1)).all()
    channel.close()
    print("PASS: explicit int contents")
