"""gRPC request assembly + error mapping
(reference: tritonclient/grpc/_utils.py)."""

import grpc

from ..utils import InferenceServerException, _reserved_params, raise_error


def get_error_grpc(rpc_error):
    """grpc.RpcError -> InferenceServerException (reference _utils.py:34-45)."""
    return InferenceServerException(
        msg=rpc_error.details(),
        status=str(rpc_error.code()),
        debug_details=rpc_error.debug_error_string(),
    )


def raise_error_grpc(rpc_error):
    raise get_error_grpc(rpc_error) from None


def get_cancelled_error(msg=None):
    return InferenceServerException(
        msg="Locally cancelled by application!" if msg is None else msg,
        status="StatusCode.CANCELLED",
    )


def _grpc_compression_type(algorithm_str):
    """Compression string -> grpc.Compression (reference _utils.py:142-154)."""
    if algorithm_str is None:
        return grpc.Compression.NoCompression
    if algorithm_str.lower() == "deflate":
        return grpc.Compression.Deflate
    if algorithm_str.lower() == "gzip":
        return grpc.Compression.Gzip
    print(
        "The provided compression algorithm is not supported. Falling back "
        "to using no compression."
    )
    return grpc.Compression.NoCompression


def _set_parameter(param, value):
    if isinstance(value, bool):
        param.bool_param = value
    elif isinstance(value, int):
        param.int64_param = value
    elif isinstance(value, str):
        param.string_param = value
    elif isinstance(value, float):
        param.double_param = value
    else:
        raise_error(
            f"unsupported parameter type {type(value)}; must be bool/int/str/float"
        )


def _get_inference_request(
    infer_request,
    model_name,
    inputs,
    model_version,
    request_id,
    outputs,
    sequence_id,
    sequence_start,
    sequence_end,
    priority,
    timeout,
    parameters,
):
    """Populate ``infer_request`` (a ModelInferRequest, possibly reused
    across calls the way the C++ client recycles its protobuf —
    grpc_client.cc:1471-1568) and return it."""
    infer_request.Clear()
    infer_request.model_name = model_name
    infer_request.model_version = model_version
    if request_id != "":
        infer_request.id = request_id
    if sequence_id != 0 and sequence_id != "":
        if isinstance(sequence_id, str):
            infer_request.parameters["sequence_id"].string_param = sequence_id
        else:
            infer_request.parameters["sequence_id"].int64_param = sequence_id
        infer_request.parameters["sequence_start"].bool_param = sequence_start
        infer_request.parameters["sequence_end"].bool_param = sequence_end
    if priority != 0:
        infer_request.parameters["priority"].uint64_param = priority
    if timeout is not None:
        infer_request.parameters["timeout"].int64_param = timeout
    for infer_input in inputs:
        infer_request.inputs.add().CopyFrom(infer_input._get_tensor())
        raw = infer_input._get_content()
        if raw is not None:
            infer_request.raw_input_contents.append(raw)
    if outputs is not None:
        for infer_output in outputs:
            infer_request.outputs.add().CopyFrom(infer_output._get_tensor())
    if parameters:
        for key, value in parameters.items():
            if key in _reserved_params:
                raise_error(
                    f"Parameter {key} is a reserved parameter and cannot be specified."
                )
            _set_parameter(infer_request.parameters[key], value)
    return infer_request
