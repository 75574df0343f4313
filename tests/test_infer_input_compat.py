"""Error-string compat contract for InferInput.set_data_from_numpy.

User code in the wild matches on these exact messages, so they are part
of the public API surface. The implementations are original (see
client_amd/{http,grpc}/_infer_input.py); the STRINGS are pinned here
against the reference wording:

- dtype mismatch:    reference http/_infer_input.py ~:139-146
- shape mismatch:    reference http/_infer_input.py ~:148-160
- BF16 JSON-path:    reference http/_infer_input.py:169-171
- UTF-8 failure:     reference http/_infer_input.py ~:186-193
"""

import numpy as np
import pytest

from client_amd import grpc as grpcclient
from client_amd import http as httpclient
from client_amd.utils import InferenceServerException


def _msg(excinfo):
    return excinfo.value.message()


def test_http_dtype_mismatch_message():
    inp = httpclient.InferInput("X", [2], "FP32")
    with pytest.raises(InferenceServerException) as e:
        inp.set_data_from_numpy(np.zeros(2, np.int32))
    assert _msg(e) == (
        "got unexpected datatype INT32 from numpy array, expected FP32"
    )


def test_http_shape_mismatch_message():
    inp = httpclient.InferInput("X", [2, 3], "FP32")
    with pytest.raises(InferenceServerException) as e:
        inp.set_data_from_numpy(np.zeros((4, 5), np.float32))
    assert _msg(e) == (
        "got unexpected numpy array shape [4, 5], expected [2, 3]"
    )


def test_http_not_ndarray_message():
    inp = httpclient.InferInput("X", [2], "FP32")
    with pytest.raises(InferenceServerException) as e:
        inp.set_data_from_numpy([1.0, 2.0])
    assert _msg(e) == "input_tensor must be a numpy array"


def test_http_bf16_json_message():
    inp = httpclient.InferInput("X", [2], "BF16")
    with pytest.raises(InferenceServerException) as e:
        inp.set_data_from_numpy(np.zeros(2, np.float32), binary_data=False)
    # reference http/_infer_input.py:169-171 verbatim
    assert _msg(e) == (
        "BF16 inputs must be sent as binary data over HTTP. "
        "Please set binary_data=True"
    )


def test_http_bytes_utf8_failure_message():
    inp = httpclient.InferInput("X", [1], "BYTES")
    bad = np.array([b"\xff\xfe"], dtype=np.object_)
    with pytest.raises(InferenceServerException) as e:
        inp.set_data_from_numpy(bad, binary_data=False)
    assert "using UTF-8" in _msg(e)
    assert "binary_data=True" in _msg(e)


def test_http_bytes_json_mixed_elements():
    # object arrays may hold bytes AND non-bytes; bytes are utf-8
    # decoded, everything else is stringified (reference nditer loop
    # behavior, http/_infer_input.py:175-193)
    inp = httpclient.InferInput("X", [3], "BYTES")
    arr = np.array([b"ab", "cd", 7], dtype=np.object_)
    inp.set_data_from_numpy(arr, binary_data=False)
    assert inp._get_tensor()["data"] == ["ab", "cd", "7"]


def test_http_fixed_width_bytes_json():
    inp = httpclient.InferInput("X", [2], "BYTES")
    arr = np.array([b"xy", b"z"], dtype="S2")
    inp.set_data_from_numpy(arr, binary_data=False)
    assert inp._get_tensor()["data"] == ["xy", "z"]


def test_grpc_dtype_mismatch_message():
    inp = grpcclient.InferInput("X", [2], "FP32")
    with pytest.raises(InferenceServerException) as e:
        inp.set_data_from_numpy(np.zeros(2, np.int64))
    assert _msg(e) == (
        "got unexpected datatype INT64 from numpy array, expected FP32"
    )


def test_grpc_shape_mismatch_message():
    inp = grpcclient.InferInput("X", [4], "FP32")
    with pytest.raises(InferenceServerException) as e:
        inp.set_data_from_numpy(np.zeros(3, np.float32))
    # the trailing comma on 1-D shapes is a reference quirk
    # (str(tuple)[1:-1] on a 1-tuple) and part of the pinned wording
    assert _msg(e) == "got unexpected numpy array shape [3,], expected [4]"


def test_numpy_data_clears_shm_binding_http():
    inp = httpclient.InferInput("X", [2], "FP32")
    inp.set_shared_memory("region", 8, offset=4)
    inp.set_data_from_numpy(np.zeros(2, np.float32))
    params = inp._get_tensor().get("parameters", {})
    assert "shared_memory_region" not in params
    assert "shared_memory_offset" not in params
    assert params.get("binary_data_size") == 8


def test_numpy_data_clears_shm_binding_grpc():
    inp = grpcclient.InferInput("X", [2], "FP32")
    inp.set_shared_memory("region", 8, offset=4)
    inp.set_data_from_numpy(np.zeros(2, np.float32))
    assert "shared_memory_region" not in inp._get_tensor().parameters
    assert inp._get_content() == b"\x00" * 8
