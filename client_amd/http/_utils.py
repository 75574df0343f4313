"""HTTP request assembly + error mapping.

Builds the KServe-v2 request body: a JSON header optionally followed by
concatenated raw tensor bytes, with the JSON length carried in the
``Inference-Header-Content-Length`` HTTP header (reference:
tritonclient/http/_utils.py:90-151).
"""

import json

from ..utils import InferenceServerException, _reserved_params, raise_error


def _raise_if_error(status, response_body):
    """Map a non-2xx HTTP response to InferenceServerException
    (reference http/_utils.py:40-74)."""
    if status >= 400:
        error_response = None
        try:
            error_response = json.loads(response_body)
        except Exception:
            pass
        if error_response is not None and "error" in error_response:
            raise InferenceServerException(
                msg=error_response["error"], status=str(status)
            )
        raise InferenceServerException(
            msg=response_body.decode("utf-8", errors="replace")
            if isinstance(response_body, (bytes, bytearray))
            else str(response_body),
            status=str(status),
        )


def _get_inference_request(
    inputs,
    request_id,
    outputs,
    sequence_id,
    sequence_start,
    sequence_end,
    priority,
    timeout,
    custom_parameters,
):
    """Return ``(request_body_bytes, json_size_or_None)``.

    json_size is None when the body is pure JSON (no trailing binary).
    """
    infer_request = {}
    parameters = {}
    if request_id != "":
        infer_request["id"] = request_id
    if sequence_id != 0 and sequence_id != "":
        parameters["sequence_id"] = sequence_id
        parameters["sequence_start"] = sequence_start
        parameters["sequence_end"] = sequence_end
    if priority != 0:
        parameters["priority"] = priority
    if timeout is not None:
        parameters["timeout"] = timeout

    infer_request["inputs"] = [this_input._get_tensor() for this_input in inputs]
    if outputs:
        infer_request["outputs"] = [
            this_output._get_tensor() for this_output in outputs
        ]
    else:
        # no outputs specified => return all outputs as binary
        parameters["binary_data_output"] = True

    if custom_parameters:
        for key, value in custom_parameters.items():
            if key in _reserved_params:
                raise_error(
                    f"Parameter {key} is a reserved parameter and cannot be specified."
                )
            parameters[key] = value
    if parameters:
        infer_request["parameters"] = parameters

    request_body = json.dumps(infer_request).encode("utf-8")
    json_size = len(request_body)

    binary_data = None
    for this_input in inputs:
        raw_data = this_input._get_binary_data()
        if raw_data is not None:
            if binary_data is None:
                binary_data = [raw_data]
            else:
                binary_data.append(raw_data)

    if binary_data is not None:
        request_body = request_body + b"".join(binary_data)
        return request_body, json_size

    return request_body, None
