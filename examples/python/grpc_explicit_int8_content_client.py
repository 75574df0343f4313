#!/usr/bin/env python3
"""INT8 tensors: no typed contents field exists for int8, so data rides
raw_input_contents as packed bytes
(reference: grpc_explicit_int8_content_client.py)."""
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
    request.model_name = "identity_int8"
    arr = np.arange(-8, 8, dtype=np.int8).reshape(1, 16)
    t = request.inputs.add()
    t.name = "INPUT0"
    t.datatype = "INT8"
    t.shape.extend(arr.shape)
    request.raw_input_contents.append(arr.tobytes())
    response = infer(request)
    out = np.frombuffer(response.raw_output_contents[0], dtype=np.int8)
    assert (out.reshape(1, 16) == arr).all()
    channel.close()
    print("PASS: explicit int8 (raw) contents")
