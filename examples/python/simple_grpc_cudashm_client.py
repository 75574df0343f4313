#!/usr/bin/env python3
"""HIP-IPC shared-memory I/O over gRPC — tensors stay in HBM3E; the
64-byte hipIpcMemHandle_t rides raw in the proto; requires 1 GPU and an
out-of-process server (reference: simple_grpc_cudashm_client.py; on
this stack cuda_shared_memory IS hip_shared_memory)."""
import argparse

import numpy as np

import tritonclient.grpc as grpcclient
import tritonclient.utils.cuda_shared_memory as cudashm

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8001")
    args = parser.parse_args()

    with grpcclient.InferenceServerClient(args.url) as client:
        client.unregister_cuda_shared_memory()
        in0 = np.arange(16, dtype=np.int32).reshape(1, 16)
        in1 = np.ones((1, 16), dtype=np.int32)
        in_handle = cudashm.create_shared_memory_region("input_data", 128, 0)
        out_handle = cudashm.create_shared_memory_region("output_data", 128, 0)
        cudashm.set_shared_memory_region(in_handle, [in0, in1])
        client.register_cuda_shared_memory(
            "input_data", cudashm.get_raw_handle_bytes(in_handle), 0, 128)
        client.register_cuda_shared_memory(
            "output_data", cudashm.get_raw_handle_bytes(out_handle), 0, 128)
        inputs = [
            grpcclient.InferInput("INPUT0", [1, 16], "INT32"),
            grpcclient.InferInput("INPUT1", [1, 16], "INT32"),
        ]
        inputs[0].set_shared_memory("input_data", 64, 0)
        inputs[1].set_shared_memory("input_data", 64, 64)
        outputs = [
            grpcclient.InferRequestedOutput("OUTPUT0"),
            grpcclient.InferRequestedOutput("OUTPUT1"),
        ]
        outputs[0].set_shared_memory("output_data", 64, 0)
        outputs[1].set_shared_memory("output_data", 64, 64)
        client.infer("simple", inputs, outputs=outputs)
        out0 = cudashm.get_contents_as_numpy(out_handle, np.int32, [1, 16], 0)
        out1 = cudashm.get_contents_as_numpy(out_handle, np.int32, [1, 16], 64)
        assert (out0 == in0 + in1).all() and (out1 == in0 - in1).all()
        client.unregister_cuda_shared_memory()
        cudashm.destroy_shared_memory_region(in_handle)
        cudashm.destroy_shared_memory_region(out_handle)
        print("PASS: grpc HIP shm")
