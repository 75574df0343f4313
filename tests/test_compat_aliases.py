"""Drop-in compatibility: user code written against tritonclient runs
unmodified; deprecation shims warn (reference tritonhttpclient shim
__init__.py:28-40)."""

import warnings

import numpy as np
import pytest


def test_tritonclient_http_alias(http_fixture_server):
    import tritonclient.http as httpclient

    host, port, _ = http_fixture_server
    client = httpclient.InferenceServerClient(f"{host}:{port}")
    try:
        assert client.is_server_live()
        x = np.random.rand(1, 5).astype(np.float32)
        inp = httpclient.InferInput("INPUT0", [1, 5], "FP32")
        inp.set_data_from_numpy(x)
        result = client.infer("identity_fp32", [inp])
        np.testing.assert_array_equal(result.as_numpy("OUTPUT0"), x)
    finally:
        client.close()


def test_tritonclient_grpc_alias(grpc_fixture_server):
    import tritonclient.grpc as grpcclient

    host, port, _ = grpc_fixture_server
    client = grpcclient.InferenceServerClient(f"{host}:{port}")
    try:
        assert client.is_server_live()
    finally:
        client.close()


def test_tritonclient_utils_alias():
    from tritonclient.utils import (
        np_to_triton_dtype,
        serialize_byte_tensor,
        InferenceServerException,
    )

    assert np_to_triton_dtype(np.float32) == "FP32"
    assert issubclass(InferenceServerException, Exception)


def test_deprecation_shims_warn():
    import importlib
    import sys

    for name in ("tritonhttpclient", "tritongrpcclient", "tritonclientutils",
                 "tritonshmutils"):
        sys.modules.pop(name, None)
        with warnings.catch_warnings(record=True) as caught:
            warnings.simplefilter("always")
            importlib.import_module(name)
        assert any(issubclass(w.category, DeprecationWarning) for w in caught), name


def test_model_config_pb2_alias():
    """Reference user code: from tritonclient.grpc import model_config_pb2."""
    from tritonclient.grpc import model_config_pb2 as mc

    cfg = mc.ModelConfig()
    cfg.name = "m"
    inp = cfg.input.add()
    inp.data_type = mc.TYPE_FP32
    assert mc.TYPE_FP32 == 11
    ig = cfg.instance_group.add()
    ig.kind = mc.ModelInstanceGroup.KIND_GPU
    data = cfg.SerializeToString()
    cfg2 = mc.ModelConfig()
    cfg2.ParseFromString(data)
    assert cfg2.input[0].data_type == mc.TYPE_FP32
    assert cfg2.instance_group[0].kind == mc.ModelInstanceGroup.KIND_GPU
    # classes are the same objects the service schema uses
    from tritonclient.grpc import service_pb2

    assert mc.ModelConfig is service_pb2.ModelConfig
