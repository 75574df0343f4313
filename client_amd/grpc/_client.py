"""KServe-v2 gRPC client.

API-compatible with ``tritonclient.grpc.InferenceServerClient``
(reference: tritonclient/grpc/_client.py). Built on grpcio generic
callables over the runtime-constructed schema in ``_proto.py`` (this
environment has no protoc). Covers sync + future-based async inference,
bi-di streaming with decoupled final-response semantics, cancellation,
keepalive, compression, and the full management surface.
"""

import grpc

from .._client import InferenceServerClientBase
from .._request import Request
from ..utils import raise_error
from ._infer_result import InferResult
from ._infer_stream import _InferStream, _RequestIterator
from ._proto import RPCS, SERVICE_NAME, service_pb2
from ._utils import (
    _get_inference_request,
    _grpc_compression_type,
    get_cancelled_error,
    get_error_grpc,
    raise_error_grpc,
)

# INT32_MAX, mirroring the reference's unlimited message sizes
# (reference common.h:53, grpc/_client.py MAX_GRPC_MESSAGE_SIZE).
MAX_GRPC_MESSAGE_SIZE = 2**31 - 1


class KeepAliveOptions:
    """gRPC keepalive knobs (reference grpc_client.h:62-82)."""

    def __init__(
        self,
        keepalive_time_ms=2**31 - 1,
        keepalive_timeout_ms=20000,
        keepalive_permit_without_calls=False,
        http2_max_pings_without_data=2,
    ):
        self.keepalive_time_ms = keepalive_time_ms
        self.keepalive_timeout_ms = keepalive_timeout_ms
        self.keepalive_permit_without_calls = keepalive_permit_without_calls
        self.http2_max_pings_without_data = http2_max_pings_without_data


class CallContext:
    """Cancellation handle returned by async_infer
    (reference grpc/_client.py:101-117)."""

    def __init__(self, grpc_future):
        self._future = grpc_future

    def cancel(self):
        return self._future.cancel()


def _get_metadata(headers, request_obj=None):
    if headers is None:
        return None
    return tuple((k.lower(), str(v)) for k, v in headers.items())


class InferenceServerClient(InferenceServerClientBase):
    def __init__(
        self,
        url,
        verbose=False,
        ssl=False,
        root_certificates=None,
        private_key=None,
        certificate_chain=None,
        creds=None,
        keepalive_options=None,
        channel_args=None,
    ):
        super().__init__()
        if keepalive_options is None:
            keepalive_options = KeepAliveOptions()
        channel_opt = [
            ("grpc.max_send_message_length", MAX_GRPC_MESSAGE_SIZE),
            ("grpc.max_receive_message_length", MAX_GRPC_MESSAGE_SIZE),
            ("grpc.keepalive_time_ms", keepalive_options.keepalive_time_ms),
            ("grpc.keepalive_timeout_ms", keepalive_options.keepalive_timeout_ms),
            (
                "grpc.keepalive_permit_without_calls",
                keepalive_options.keepalive_permit_without_calls,
            ),
            (
                "grpc.http2.max_pings_without_data",
                keepalive_options.http2_max_pings_without_data,
            ),
        ]
        if channel_args is not None:
            channel_opt.extend(channel_args)
        if creds is not None:
            self._channel = grpc.secure_channel(url, creds, options=channel_opt)
        elif ssl:
            rc = pk = cc = None
            if root_certificates is not None:
                with open(root_certificates, "rb") as f:
                    rc = f.read()
            if private_key is not None:
                with open(private_key, "rb") as f:
                    pk = f.read()
            if certificate_chain is not None:
                with open(certificate_chain, "rb") as f:
                    cc = f.read()
            credentials = grpc.ssl_channel_credentials(rc, pk, cc)
            self._channel = grpc.secure_channel(url, credentials, options=channel_opt)
        else:
            self._channel = grpc.insecure_channel(url, options=channel_opt)
        self._verbose = verbose
        self._stream = None
        # Build one callable per RPC from the runtime schema.
        self._rpc = {}
        for name, (req_cls, resp_cls, streaming) in RPCS.items():
            path = f"/{SERVICE_NAME}/{name}"
            if streaming:
                self._rpc[name] = self._channel.stream_stream(
                    path,
                    request_serializer=req_cls.SerializeToString,
                    response_deserializer=resp_cls.FromString,
                )
            else:
                self._rpc[name] = self._channel.unary_unary(
                    path,
                    request_serializer=req_cls.SerializeToString,
                    response_deserializer=resp_cls.FromString,
                )
        # Reused request protobuf for sync infer (reference
        # grpc_client.cc:1471-1530 recycles submessages the same way).
        self._infer_request = service_pb2.ModelInferRequest()

    def __enter__(self):
        return self

    def __exit__(self, type, value, traceback):
        self.close()

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def close(self):
        self.stop_stream()
        self._channel.close()

    def _metadata(self, headers):
        request = Request(dict(headers) if headers else {})
        self._call_plugin(request)
        if not request.headers:
            return None
        return tuple((k.lower(), str(v)) for k, v in request.headers.items())

    # ---- health / metadata ----

    def is_server_live(self, headers=None, client_timeout=None):
        try:
            request = service_pb2.ServerLiveRequest()
            response = self._rpc["ServerLive"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            return response.live
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def is_server_ready(self, headers=None, client_timeout=None):
        try:
            request = service_pb2.ServerReadyRequest()
            response = self._rpc["ServerReady"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            return response.ready
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def is_model_ready(
        self, model_name, model_version="", headers=None, client_timeout=None
    ):
        try:
            request = service_pb2.ModelReadyRequest(
                name=model_name, version=model_version
            )
            response = self._rpc["ModelReady"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            return response.ready
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def get_server_metadata(self, headers=None, as_json=False, client_timeout=None):
        try:
            request = service_pb2.ServerMetadataRequest()
            response = self._rpc["ServerMetadata"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            return self._return(response, as_json)
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def get_model_metadata(
        self, model_name, model_version="", headers=None, as_json=False,
        client_timeout=None,
    ):
        try:
            request = service_pb2.ModelMetadataRequest(
                name=model_name, version=model_version
            )
            response = self._rpc["ModelMetadata"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            return self._return(response, as_json)
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def get_model_config(
        self, model_name, model_version="", headers=None, as_json=False,
        client_timeout=None,
    ):
        try:
            request = service_pb2.ModelConfigRequest(
                name=model_name, version=model_version
            )
            response = self._rpc["ModelConfig"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            return self._return(response, as_json)
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    @staticmethod
    def _return(response, as_json):
        if as_json:
            from google.protobuf.json_format import MessageToDict

            return MessageToDict(response, preserving_proto_field_name=True)
        return response

    # ---- repository ----

    def get_model_repository_index(self, headers=None, as_json=False,
                                   client_timeout=None):
        try:
            request = service_pb2.RepositoryIndexRequest()
            response = self._rpc["RepositoryIndex"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            return self._return(response, as_json)
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def load_model(
        self, model_name, headers=None, config=None, files=None, client_timeout=None
    ):
        try:
            request = service_pb2.RepositoryModelLoadRequest(model_name=model_name)
            if config is not None:
                request.parameters["config"].string_param = config
            if files is not None:
                for path, content in files.items():
                    request.parameters[path].bytes_param = content
            self._rpc["RepositoryModelLoad"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            if self._verbose:
                print("Loaded model '{}'".format(model_name))
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def unload_model(
        self, model_name, headers=None, unload_dependents=False, client_timeout=None
    ):
        try:
            request = service_pb2.RepositoryModelUnloadRequest(model_name=model_name)
            request.parameters["unload_dependents"].bool_param = unload_dependents
            self._rpc["RepositoryModelUnload"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            if self._verbose:
                print("Unloaded model '{}'".format(model_name))
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    # ---- statistics / trace / log ----

    def get_inference_statistics(
        self, model_name="", model_version="", headers=None, as_json=False,
        client_timeout=None,
    ):
        try:
            request = service_pb2.ModelStatisticsRequest(
                name=model_name, version=model_version
            )
            response = self._rpc["ModelStatistics"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            return self._return(response, as_json)
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def update_trace_settings(
        self, model_name=None, settings={}, headers=None, as_json=False,
        client_timeout=None,
    ):
        try:
            request = service_pb2.TraceSettingRequest()
            if model_name is not None:
                request.model_name = model_name
            for key, value in settings.items():
                if value is not None:
                    if isinstance(value, (list, tuple)):
                        request.settings[key].value.extend([str(v) for v in value])
                    else:
                        request.settings[key].value.extend([str(value)])
                else:
                    request.settings[key]
            response = self._rpc["TraceSetting"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            return self._return(response, as_json)
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def get_trace_settings(self, model_name=None, headers=None, as_json=False,
                           client_timeout=None):
        return self.update_trace_settings(
            model_name=model_name, settings={}, headers=headers, as_json=as_json,
            client_timeout=client_timeout,
        )

    def update_log_settings(self, settings, headers=None, as_json=False,
                            client_timeout=None):
        try:
            request = service_pb2.LogSettingsRequest()
            for key, value in settings.items():
                if value is not None:
                    if isinstance(value, bool):
                        request.settings[key].bool_param = value
                    elif isinstance(value, int):
                        request.settings[key].uint32_param = value
                    elif isinstance(value, str):
                        request.settings[key].string_param = value
                    else:
                        raise_error(f"Unsupported log setting type {type(value)}")
                else:
                    request.settings[key]
            response = self._rpc["LogSettings"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            return self._return(response, as_json)
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def get_log_settings(self, headers=None, as_json=False, client_timeout=None):
        return self.update_log_settings(
            settings={}, headers=headers, as_json=as_json,
            client_timeout=client_timeout,
        )

    # ---- shared memory ----

    def get_system_shared_memory_status(
        self, region_name="", headers=None, as_json=False, client_timeout=None
    ):
        try:
            request = service_pb2.SystemSharedMemoryStatusRequest(name=region_name)
            response = self._rpc["SystemSharedMemoryStatus"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            return self._return(response, as_json)
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def register_system_shared_memory(
        self, name, key, byte_size, offset=0, headers=None, client_timeout=None
    ):
        try:
            request = service_pb2.SystemSharedMemoryRegisterRequest(
                name=name, key=key, offset=offset, byte_size=byte_size
            )
            self._rpc["SystemSharedMemoryRegister"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            if self._verbose:
                print("Registered system shared memory with name '{}'".format(name))
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def unregister_system_shared_memory(self, name="", headers=None,
                                        client_timeout=None):
        try:
            request = service_pb2.SystemSharedMemoryUnregisterRequest(name=name)
            self._rpc["SystemSharedMemoryUnregister"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            if self._verbose:
                if name != "":
                    print(
                        "Unregistered system shared memory with name '{}'".format(name)
                    )
                else:
                    print("Unregistered all system shared memory regions")
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def get_cuda_shared_memory_status(
        self, region_name="", headers=None, as_json=False, client_timeout=None
    ):
        try:
            request = service_pb2.CudaSharedMemoryStatusRequest(name=region_name)
            response = self._rpc["CudaSharedMemoryStatus"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            return self._return(response, as_json)
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def register_cuda_shared_memory(
        self, name, raw_handle, device_id, byte_size, headers=None,
        client_timeout=None,
    ):
        """``raw_handle``: the 64 raw bytes of the hipIpcMemHandle_t
        (the proto carries raw bytes, not base64 —
        grpc_service.proto:1610-1643)."""
        try:
            request = service_pb2.CudaSharedMemoryRegisterRequest(
                name=name, raw_handle=raw_handle, device_id=device_id,
                byte_size=byte_size,
            )
            self._rpc["CudaSharedMemoryRegister"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            if self._verbose:
                print("Registered cuda shared memory with name '{}'".format(name))
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    register_hip_shared_memory = register_cuda_shared_memory
    get_hip_shared_memory_status = get_cuda_shared_memory_status

    def unregister_cuda_shared_memory(self, name="", headers=None,
                                      client_timeout=None):
        try:
            request = service_pb2.CudaSharedMemoryUnregisterRequest(name=name)
            self._rpc["CudaSharedMemoryUnregister"](
                request, metadata=self._metadata(headers), timeout=client_timeout
            )
            if self._verbose:
                if name != "":
                    print(
                        "Unregistered cuda shared memory with name '{}'".format(name)
                    )
                else:
                    print("Unregistered all cuda shared memory regions")
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    unregister_hip_shared_memory = unregister_cuda_shared_memory

    # ---- inference ----

    def infer(
        self,
        model_name,
        inputs,
        model_version="",
        outputs=None,
        request_id="",
        sequence_id=0,
        sequence_start=False,
        sequence_end=False,
        priority=0,
        timeout=None,
        client_timeout=None,
        headers=None,
        compression_algorithm=None,
        parameters=None,
    ):
        metadata = self._metadata(headers)
        request = _get_inference_request(
            self._infer_request,
            model_name=model_name,
            inputs=inputs,
            model_version=model_version,
            request_id=request_id,
            outputs=outputs,
            sequence_id=sequence_id,
            sequence_start=sequence_start,
            sequence_end=sequence_end,
            priority=priority,
            timeout=timeout,
            parameters=parameters,
        )
        if self._verbose:
            print("infer, metadata {}".format(metadata))
        try:
            response = self._rpc["ModelInfer"](
                request,
                metadata=metadata,
                timeout=client_timeout,
                compression=_grpc_compression_type(compression_algorithm),
            )
            return InferResult(response)
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    def async_infer(
        self,
        model_name,
        inputs,
        callback,
        model_version="",
        outputs=None,
        request_id="",
        sequence_id=0,
        sequence_start=False,
        sequence_end=False,
        priority=0,
        timeout=None,
        client_timeout=None,
        headers=None,
        compression_algorithm=None,
        parameters=None,
    ):
        """Future-based async inference; ``callback(result, error)`` fires
        on completion. Returns a CallContext with cancel()."""
        metadata = self._metadata(headers)
        # async path gets its own request object (it outlives this call)
        request = _get_inference_request(
            service_pb2.ModelInferRequest(),
            model_name=model_name,
            inputs=inputs,
            model_version=model_version,
            request_id=request_id,
            outputs=outputs,
            sequence_id=sequence_id,
            sequence_start=sequence_start,
            sequence_end=sequence_end,
            priority=priority,
            timeout=timeout,
            parameters=parameters,
        )

        def wrapped_callback(call_future):
            error = result = None
            try:
                response = call_future.result()
                result = InferResult(response)
            except grpc.RpcError as rpc_error:
                error = get_error_grpc(rpc_error)
            except grpc.FutureCancelledError:
                error = get_cancelled_error()
            callback(result=result, error=error)

        try:
            future = self._rpc["ModelInfer"].future(
                request,
                metadata=metadata,
                timeout=client_timeout,
                compression=_grpc_compression_type(compression_algorithm),
            )
            future.add_done_callback(wrapped_callback)
            return CallContext(future)
        except grpc.RpcError as rpc_error:
            raise_error_grpc(rpc_error)

    # ---- streaming ----

    def start_stream(
        self,
        callback,
        stream_timeout=None,
        headers=None,
        compression_algorithm=None,
    ):
        """Open the bi-di ModelStreamInfer stream; responses (including
        decoupled multi-response models) are delivered to ``callback``
        on a dedicated reader thread (reference grpc/_client.py:1743-1798)."""
        if self._stream is not None:
            raise_error(
                "cannot start another stream with one already running. "
                "'InferenceServerClient' supports only a single active "
                "stream at a given time."
            )
        metadata = self._metadata(headers)
        self._stream = _InferStream(callback, self._verbose)
        try:
            response_iterator = self._rpc["ModelStreamInfer"](
                _RequestIterator(self._stream),
                metadata=metadata,
                timeout=stream_timeout,
                compression=_grpc_compression_type(compression_algorithm),
            )
            self._stream._init_handler(response_iterator)
        except grpc.RpcError as rpc_error:
            self._stream = None
            raise_error_grpc(rpc_error)

    def stop_stream(self, cancel_requests=False):
        if self._stream is not None:
            self._stream.close(cancel_requests)
            self._stream = None

    def async_stream_infer(
        self,
        model_name,
        inputs,
        model_version="",
        outputs=None,
        request_id="",
        sequence_id=0,
        sequence_start=False,
        sequence_end=False,
        enable_empty_final_response=False,
        priority=0,
        timeout=None,
        parameters=None,
    ):
        """Enqueue one request onto the open stream."""
        if self._stream is None:
            raise_error("stream not available, use start_stream() to make one")
        request = _get_inference_request(
            service_pb2.ModelInferRequest(),
            model_name=model_name,
            inputs=inputs,
            model_version=model_version,
            request_id=request_id,
            outputs=outputs,
            sequence_id=sequence_id,
            sequence_start=sequence_start,
            sequence_end=sequence_end,
            priority=priority,
            timeout=timeout,
            parameters=parameters,
        )
        if enable_empty_final_response:
            request.parameters["triton_enable_empty_final_response"].bool_param = True
        if self._verbose:
            print("async_stream_infer")
        self._stream._enqueue_request(request)
