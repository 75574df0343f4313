"""client_amd.grpc.aio — asyncio KServe-v2 gRPC client on grpc.aio.

Mirrors tritonclient.grpc.aio (reference:
tritonclient/grpc/aio/__init__.py — coroutine management + infer
:634-686, stream_infer returning a cancellable response iterator
:688-809).
"""

import grpc

from ..._client import InferenceServerClientBase
from ..._request import Request
from ...utils import InferenceServerException, raise_error
from .._infer_input import InferInput  # re-export for API parity
from .._infer_result import InferResult
from .._proto import RPCS, SERVICE_NAME, service_pb2
from .._requested_output import InferRequestedOutput  # re-export
from .._client import KeepAliveOptions, MAX_GRPC_MESSAGE_SIZE
from .._utils import (
    _get_inference_request,
    _grpc_compression_type,
    get_cancelled_error,
    get_error_grpc,
    raise_error_grpc,
)

__all__ = [
    "InferenceServerClient",
    "InferInput",
    "InferRequestedOutput",
    "InferResult",
    "InferenceServerException",
    "KeepAliveOptions",
]


class InferenceServerClient(InferenceServerClientBase):
    def __init__(self, url, verbose=False, ssl=False, root_certificates=None,
                 private_key=None, certificate_chain=None, creds=None,
                 keepalive_options=None, channel_args=None):
        super().__init__()
        if keepalive_options is None:
            keepalive_options = KeepAliveOptions()
        channel_opt = [
            ("grpc.max_send_message_length", MAX_GRPC_MESSAGE_SIZE),
            ("grpc.max_receive_message_length", MAX_GRPC_MESSAGE_SIZE),
            ("grpc.keepalive_time_ms", keepalive_options.keepalive_time_ms),
            ("grpc.keepalive_timeout_ms", keepalive_options.keepalive_timeout_ms),
            ("grpc.keepalive_permit_without_calls",
             keepalive_options.keepalive_permit_without_calls),
            ("grpc.http2.max_pings_without_data",
             keepalive_options.http2_max_pings_without_data),
        ]
        if channel_args is not None:
            channel_opt.extend(channel_args)
        if creds is not None:
            self._channel = grpc.aio.secure_channel(url, creds, options=channel_opt)
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
            self._channel = grpc.aio.secure_channel(url, credentials,
                                                    options=channel_opt)
        else:
            self._channel = grpc.aio.insecure_channel(url, options=channel_opt)
        self._verbose = verbose
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

    async def __aenter__(self):
        return self

    async def __aexit__(self, type, value, traceback):
        await self.close()

    async def close(self):
        await self._channel.close()

    def _metadata(self, headers):
        request = Request(dict(headers) if headers else {})
        self._call_plugin(request)
        if not request.headers:
            return None
        return tuple((k.lower(), str(v)) for k, v in request.headers.items())

    @staticmethod
    def _return(response, as_json):
        if as_json:
            from google.protobuf.json_format import MessageToDict

            return MessageToDict(response, preserving_proto_field_name=True)
        return response

    # ---- health / metadata ----

    async def is_server_live(self, headers=None):
        try:
            response = await self._rpc["ServerLive"](
                service_pb2.ServerLiveRequest(), metadata=self._metadata(headers)
            )
            return response.live
        except grpc.RpcError as e:
            raise_error_grpc(e)

    async def is_server_ready(self, headers=None):
        try:
            response = await self._rpc["ServerReady"](
                service_pb2.ServerReadyRequest(), metadata=self._metadata(headers)
            )
            return response.ready
        except grpc.RpcError as e:
            raise_error_grpc(e)

    async def is_model_ready(self, model_name, model_version="", headers=None):
        try:
            response = await self._rpc["ModelReady"](
                service_pb2.ModelReadyRequest(name=model_name,
                                              version=model_version),
                metadata=self._metadata(headers),
            )
            return response.ready
        except grpc.RpcError as e:
            raise_error_grpc(e)

    async def get_server_metadata(self, headers=None, as_json=False):
        try:
            response = await self._rpc["ServerMetadata"](
                service_pb2.ServerMetadataRequest(), metadata=self._metadata(headers)
            )
            return self._return(response, as_json)
        except grpc.RpcError as e:
            raise_error_grpc(e)

    async def get_model_metadata(self, model_name, model_version="", headers=None,
                                 as_json=False):
        try:
            response = await self._rpc["ModelMetadata"](
                service_pb2.ModelMetadataRequest(name=model_name,
                                                 version=model_version),
                metadata=self._metadata(headers),
            )
            return self._return(response, as_json)
        except grpc.RpcError as e:
            raise_error_grpc(e)

    async def get_model_config(self, model_name, model_version="", headers=None,
                               as_json=False):
        try:
            response = await self._rpc["ModelConfig"](
                service_pb2.ModelConfigRequest(name=model_name,
                                               version=model_version),
                metadata=self._metadata(headers),
            )
            return self._return(response, as_json)
        except grpc.RpcError as e:
            raise_error_grpc(e)

    async def get_model_repository_index(self, headers=None, as_json=False):
        try:
            response = await self._rpc["RepositoryIndex"](
                service_pb2.RepositoryIndexRequest(),
                metadata=self._metadata(headers),
            )
            return self._return(response, as_json)
        except grpc.RpcError as e:
            raise_error_grpc(e)

    async def load_model(self, model_name, headers=None, config=None, files=None):
        try:
            request = service_pb2.RepositoryModelLoadRequest(model_name=model_name)
            if config is not None:
                request.parameters["config"].string_param = config
            if files is not None:
                for path, content in files.items():
                    request.parameters[path].bytes_param = content
            await self._rpc["RepositoryModelLoad"](
                request, metadata=self._metadata(headers)
            )
        except grpc.RpcError as e:
            raise_error_grpc(e)

    async def unload_model(self, model_name, headers=None,
                           unload_dependents=False):
        try:
            request = service_pb2.RepositoryModelUnloadRequest(
                model_name=model_name
            )
            request.parameters["unload_dependents"].bool_param = unload_dependents
            await self._rpc["RepositoryModelUnload"](
                request, metadata=self._metadata(headers)
            )
        except grpc.RpcError as e:
            raise_error_grpc(e)

    async def get_inference_statistics(self, model_name="", model_version="",
                                       headers=None, as_json=False):
        try:
            response = await self._rpc["ModelStatistics"](
                service_pb2.ModelStatisticsRequest(name=model_name,
                                                   version=model_version),
                metadata=self._metadata(headers),
            )
            return self._return(response, as_json)
        except grpc.RpcError as e:
            raise_error_grpc(e)

    # ---- inference ----

    async def infer(
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
        try:
            response = await self._rpc["ModelInfer"](
                request,
                metadata=metadata,
                timeout=client_timeout,
                compression=_grpc_compression_type(compression_algorithm),
            )
            return InferResult(response)
        except grpc.RpcError as e:
            raise_error_grpc(e)

    def stream_infer(self, inputs_iterator, stream_timeout=None, headers=None,
                     compression_algorithm=None):
        """Bi-di streaming: ``inputs_iterator`` is an async iterator of
        request kwargs dicts (same keys as infer()); returns an async
        iterator of (InferResult, error) tuples with .cancel()
        (reference grpc/aio/__init__.py:688-809)."""
        metadata = self._metadata(headers)

        async def _request_iterator():
            async for kwargs in inputs_iterator:
                request = _get_inference_request(
                    service_pb2.ModelInferRequest(),
                    model_name=kwargs["model_name"],
                    inputs=kwargs["inputs"],
                    model_version=kwargs.get("model_version", ""),
                    request_id=kwargs.get("request_id", ""),
                    outputs=kwargs.get("outputs"),
                    sequence_id=kwargs.get("sequence_id", 0),
                    sequence_start=kwargs.get("sequence_start", False),
                    sequence_end=kwargs.get("sequence_end", False),
                    priority=kwargs.get("priority", 0),
                    timeout=kwargs.get("timeout"),
                    parameters=kwargs.get("parameters"),
                )
                if kwargs.get("enable_empty_final_response"):
                    request.parameters[
                        "triton_enable_empty_final_response"
                    ].bool_param = True
                yield request

        call = self._rpc["ModelStreamInfer"](
            _request_iterator(),
            metadata=metadata,
            timeout=stream_timeout,
            compression=_grpc_compression_type(compression_algorithm),
        )

        class _ResponseIterator:
            def __init__(self, call):
                self._call = call

            def __aiter__(self):
                return self

            async def __anext__(self):
                try:
                    response = await self._call.read()
                except grpc.RpcError as e:
                    raise get_error_grpc(e) from None
                if response == grpc.aio.EOF:
                    raise StopAsyncIteration
                if response.error_message != "":
                    return None, InferenceServerException(
                        msg=response.error_message
                    )
                return InferResult(response.infer_response), None

            def cancel(self):
                return self._call.cancel()

        return _ResponseIterator(call)
