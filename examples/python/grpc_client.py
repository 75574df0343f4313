#!/usr/bin/env python3
"""Bare-stub gRPC usage: drives GRPCInferenceService with grpcio and the
runtime-built message classes directly, no tritonclient wrapper — the
analog of the reference's generated-stub grpc_client.py."""
import argparse

import grpc
import numpy as np

from client_amd.grpc._proto import RPCS, SERVICE_NAME, service_pb2

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()

    channel = grpc.insecure_channel(args.url)

    def rpc(name, request):
        req_cls, resp_cls, _ = RPCS[name]
        call = channel.unary_unary(
            f"/{SERVICE_NAME}/{name}",
            request_serializer=req_cls.SerializeToString,
            response_deserializer=resp_cls.FromString,
        )
        return call(request)

    live = rpc("ServerLive", service_pb2.ServerLiveRequest())
    assert live.live
    meta = rpc("ServerMetadata", service_pb2.ServerMetadataRequest())
    print("server:", meta.name, meta.version)

    request = service_pb2.ModelInferRequest()
    request.model_name = "simple"
    in0 = np.arange(16, dtype=np.int32).reshape(1, 16)
    in1 = np.ones((1, 16), dtype=np.int32)
    for name, arr in (("INPUT0", in0), ("INPUT1", in1)):
        t = request.inputs.add()
        t.name = name
        t.datatype = "INT32"
        t.shape.extend(arr.shape)
        request.raw_input_contents.append(arr.tobytes())
    response = rpc("ModelInfer", request)
    out0 = np.frombuffer(response.raw_output_contents[0], dtype=np.int32)
    assert (out0.reshape(1, 16) == in0 + in1).all()
    channel.close()
    print("PASS: bare-stub grpc client")
