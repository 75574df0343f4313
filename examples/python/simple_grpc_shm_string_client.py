#!/usr/bin/env python3
"""BYTES tensors staged through system shared memory over gRPC: the
serialized string tensor (4-byte-length-prefixed elements) lives in the
region; the request carries only region references
(reference: simple_grpc_shm_string_client.py)."""
import argparse

import numpy as np

import tritonclient.grpc as grpcclient
import tritonclient.utils.shared_memory as shm
from tritonclient.utils import serialized_byte_size

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()

    with grpcclient.InferenceServerClient(args.url) as client:
        client.unregister_system_shared_memory()
        in0 = np.array([str(i).encode() for i in range(16)],
                       dtype=np.object_).reshape(1, 16)
        in1 = np.array([b"1"] * 16, dtype=np.object_).reshape(1, 16)
        # region is sized from the serialized (4-byte-length-prefixed)
        # form; set_shared_memory_region serializes object arrays itself
        n0 = serialized_byte_size(in0)
        n1 = serialized_byte_size(in1)
        nbytes = n0 + n1
        handle = shm.create_shared_memory_region(
            "str_io", "/simple_grpc_shm_string", nbytes)
        shm.set_shared_memory_region(handle, [in0, in1])
        client.register_system_shared_memory(
            "str_io", "/simple_grpc_shm_string", nbytes)
        inputs = [
            grpcclient.InferInput("INPUT0", [1, 16], "BYTES"),
            grpcclient.InferInput("INPUT1", [1, 16], "BYTES"),
        ]
        inputs[0].set_shared_memory("str_io", n0, 0)
        inputs[1].set_shared_memory("str_io", n1, n0)
        result = client.infer("simple_string", inputs)
        got = np.array([int(v) for v in result.as_numpy("OUTPUT0").reshape(-1)])
        assert (got == np.arange(16) + 1).all()
        client.unregister_system_shared_memory()
        shm.destroy_shared_memory_region(handle)
        print("PASS: grpc shm string")
