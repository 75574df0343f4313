#!/usr/bin/env python3
"""System shared-memory I/O: no tensor bytes on the wire
(reference: simple_http_shm_client.py)."""
import argparse

import numpy as np

import tritonclient.http as httpclient
import tritonclient.utils.shared_memory as shm

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-u", "--url", default="127.0.0.1:8000")
    args = parser.parse_args()

    with httpclient.InferenceServerClient(args.url) as client:
        client.unregister_system_shared_memory()
        in0 = np.arange(16, dtype=np.int32).reshape(1, 16)
        in1 = np.ones((1, 16), dtype=np.int32)
        shm_handle = shm.create_shared_memory_region("io", "/simple_http_shm", 256)
        shm.set_shared_memory_region(shm_handle, [in0, in1])
        client.register_system_shared_memory("io", "/simple_http_shm", 256)
        inputs = [
            httpclient.InferInput("INPUT0", [1, 16], "INT32"),
            httpclient.InferInput("INPUT1", [1, 16], "INT32"),
        ]
        inputs[0].set_shared_memory("io", 64, 0)
        inputs[1].set_shared_memory("io", 64, 64)
        outputs = [
            httpclient.InferRequestedOutput("OUTPUT0"),
            httpclient.InferRequestedOutput("OUTPUT1"),
        ]
        outputs[0].set_shared_memory("io", 64, 128)
        outputs[1].set_shared_memory("io", 64, 192)
        client.infer("simple", inputs, outputs=outputs)
        out0 = shm.get_contents_as_numpy(shm_handle, np.int32, [1, 16], 128)
        out1 = shm.get_contents_as_numpy(shm_handle, np.int32, [1, 16], 192)
        assert (out0 == in0 + in1).all() and (out1 == in0 - in1).all()
        client.unregister_system_shared_memory()
        shm.destroy_shared_memory_region(shm_handle)
        print("PASS: system shared memory")
