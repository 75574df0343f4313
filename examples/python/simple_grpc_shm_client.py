#!/usr/bin/env python3
"""System shared-memory I/O over gRPC: no tensor bytes on the wire
(reference: simple_grpc_shm_client.py)."""
import argparse

import numpy as np

import tritonclient.grpc as grpcclient
import tritonclient.utils.shared_memory as shm

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()

    with grpcclient.InferenceServerClient(args.url) as client:
        client.unregister_system_shared_memory()
        in0 = np.arange(16, dtype=np.int32).reshape(1, 16)
        in1 = np.ones((1, 16), dtype=np.int32)
        handle = shm.create_shared_memory_region("io", "/simple_grpc_shm", 256)
        shm.set_shared_memory_region(handle, [in0, in1])
        client.register_system_shared_memory("io", "/simple_grpc_shm", 256)
        inputs = [
            grpcclient.InferInput("INPUT0", [1, 16], "INT32"),
            grpcclient.InferInput("INPUT1", [1, 16], "INT32"),
        ]
        inputs[0].set_shared_memory("io", 64, 0)
        inputs[1].set_shared_memory("io", 64, 64)
        outputs = [
            grpcclient.InferRequestedOutput("OUTPUT0"),
            grpcclient.InferRequestedOutput("OUTPUT1"),
        ]
        outputs[0].set_shared_memory("io", 64, 128)
        outputs[1].set_shared_memory("io", 64, 192)
        client.infer("simple", inputs, outputs=outputs)
        out0 = shm.get_contents_as_numpy(handle, np.int32, [1, 16], 128)
        out1 = shm.get_contents_as_numpy(handle, np.int32, [1, 16], 192)
        assert (out0 == in0 + in1).all() and (out1 == in0 - in1).all()
        client.unregister_system_shared_memory()
        shm.destroy_shared_memory_region(handle)
        print("PASS: grpc system shm")
