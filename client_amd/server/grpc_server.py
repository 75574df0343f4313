"""KServe-v2 gRPC server frontend.

grpcio generic-handler service over InferenceCore, speaking the
runtime-built schema from client_amd.grpc._proto. Implements all 20
RPCs of GRPCInferenceService (reference service definition:
grpc_service.proto:40-219) including bi-di ModelStreamInfer with
decoupled multi-response models and triton_final_response semantics.
"""

from concurrent import futures

import grpc

from ..grpc._proto import RPCS, SERVICE_NAME, service_pb2
from .core import InferenceCore, InferenceError

_DT_ENUM = {
    "BOOL": 1, "UINT8": 2, "UINT16": 3, "UINT32": 4, "UINT64": 5,
    "INT8": 6, "INT16": 7, "INT32": 8, "INT64": 9, "FP16": 10,
    "FP32": 11, "FP64": 12, "BYTES": 13, "BF16": 14,
}


def _param_value(p):
    which = p.WhichOneof("parameter_choice")
    return getattr(p, which) if which else None


def _params_to_dict(pb_map):
    return {k: _param_value(v) for k, v in pb_map.items()}


class GrpcServer:
    def __init__(self, core=None, host="127.0.0.1", port=8001, max_workers=8,
                 ssl_credentials=None):
        """ssl_credentials: a grpc.ssl_server_credentials(...) object to
        serve TLS (ALPN h2) instead of plaintext h2c."""
        self.core = core if core is not None else InferenceCore()
        self.host = host
        self.port = port
        self._server = grpc.server(
            futures.ThreadPoolExecutor(max_workers=max_workers),
            options=[
                ("grpc.max_send_message_length", 2**31 - 1),
                ("grpc.max_receive_message_length", 2**31 - 1),
            ],
        )
        self._server.add_generic_rpc_handlers((self._make_handler(),))
        if ssl_credentials is not None:
            self.port = self._server.add_secure_port(
                f"{host}:{port}", ssl_credentials)
        else:
            self.port = self._server.add_insecure_port(f"{host}:{port}")

    # ---- request translation ----

    def _request_to_dict(self, request):
        """ModelInferRequest -> (core request dict, per-input raw list)."""
        req = {"parameters": _params_to_dict(request.parameters)}
        if request.id:
            req["id"] = request.id
        inputs = []
        raw_list = list(request.raw_input_contents)
        raw_iter = 0
        for inp in request.inputs:
            d = {
                "name": inp.name,
                "datatype": inp.datatype,
                "shape": list(inp.shape),
                "parameters": _params_to_dict(inp.parameters),
            }
            if "shared_memory_region" not in d["parameters"]:
                if raw_iter < len(raw_list):
                    d["parameters"]["binary_data_size"] = len(raw_list[raw_iter])
                    raw_iter += 1
                else:
                    # typed contents path
                    c = inp.contents
                    for field in ("bool_contents", "int_contents", "int64_contents",
                                  "uint_contents", "uint64_contents", "fp32_contents",
                                  "fp64_contents", "bytes_contents"):
                        vals = getattr(c, field)
                        if len(vals):
                            d["data"] = list(vals)
                            break
            inputs.append(d)
        req["inputs"] = inputs
        outputs = []
        for out in request.outputs:
            params = _params_to_dict(out.parameters)
            if "shared_memory_region" not in params:
                params["binary_data"] = True
            outputs.append({"name": out.name, "parameters": params})
        if outputs:
            req["outputs"] = outputs
        else:
            req.setdefault("parameters", {})["binary_data_output"] = True
        return req, raw_list

    def _dict_to_response(self, response_dict, binary_parts, request_id=""):
        resp = service_pb2.ModelInferResponse()
        resp.model_name = response_dict.get("model_name", "")
        resp.model_version = response_dict.get("model_version", "1")
        if response_dict.get("id"):
            resp.id = response_dict["id"]
        elif request_id:
            resp.id = request_id
        part = 0
        for out in response_dict.get("outputs", []):
            o = resp.outputs.add()
            o.name = out["name"]
            o.datatype = out["datatype"]
            o.shape.extend(out["shape"])
            params = out.get("parameters", {})
            if "binary_data_size" in params:
                resp.raw_output_contents.append(binary_parts[part])
                part += 1
            elif "shared_memory_region" in params:
                o.parameters["shared_memory_region"].string_param = params[
                    "shared_memory_region"
                ]
                o.parameters["shared_memory_byte_size"].int64_param = params[
                    "shared_memory_byte_size"
                ]
                if "shared_memory_offset" in params:
                    o.parameters["shared_memory_offset"].int64_param = params[
                        "shared_memory_offset"
                    ]
            elif "data" in out:
                data = out["data"]
                dt = out["datatype"]
                c = o.contents
                if dt == "BOOL":
                    c.bool_contents.extend(data)
                elif dt in ("INT8", "INT16", "INT32"):
                    c.int_contents.extend(data)
                elif dt == "INT64":
                    c.int64_contents.extend(data)
                elif dt in ("UINT8", "UINT16", "UINT32"):
                    c.uint_contents.extend(data)
                elif dt == "UINT64":
                    c.uint64_contents.extend(data)
                elif dt == "FP32":
                    c.fp32_contents.extend(data)
                elif dt == "FP64":
                    c.fp64_contents.extend(data)
                elif dt == "BYTES":
                    c.bytes_contents.extend(
                        v.encode() if isinstance(v, str) else v for v in data
                    )
        return resp

    # ---- RPC implementations ----

    def ServerLive(self, request, context):
        return service_pb2.ServerLiveResponse(live=self.core.live)

    def ServerReady(self, request, context):
        return service_pb2.ServerReadyResponse(ready=self.core.ready)

    def ModelReady(self, request, context):
        ready = self.core.model_state.get(request.name) == "READY"
        return service_pb2.ModelReadyResponse(ready=ready)

    def ServerMetadata(self, request, context):
        return service_pb2.ServerMetadataResponse(
            name=self.core.server_name,
            version=self.core.version,
            extensions=[
                "classification", "sequence", "model_repository",
                "model_configuration", "system_shared_memory",
                "cuda_shared_memory", "binary_tensor_data", "statistics",
                "trace", "logging",
            ],
        )

    def ModelMetadata(self, request, context):
        model = self.core.get_model(request.name, must_be_ready=False)
        meta = model.metadata()
        resp = service_pb2.ModelMetadataResponse(
            name=meta["name"], versions=meta["versions"], platform=meta["platform"]
        )
        for io_list, target in ((meta["inputs"], resp.inputs),
                                (meta["outputs"], resp.outputs)):
            for io in io_list:
                t = target.add()
                t.name = io["name"]
                t.datatype = io["datatype"]
                t.shape.extend(io["shape"])
        return resp

    def ModelConfig(self, request, context):
        model = self.core.get_model(request.name, must_be_ready=False)
        resp = service_pb2.ModelConfigResponse()
        cfg = resp.config
        cfg.name = model.name
        cfg.platform = model.platform
        cfg.backend = model.platform
        cfg.max_batch_size = model.max_batch_size
        for n, d, s in model.inputs:
            i = cfg.input.add()
            i.name = n
            i.data_type = _DT_ENUM.get(d, 0)
            i.dims.extend(s)
        for n, d, s in model.outputs:
            o = cfg.output.add()
            o.name = n
            o.data_type = _DT_ENUM.get(d, 0)
            o.dims.extend(s)
        cfg.model_transaction_policy.decoupled = model.decoupled
        # load-time config override is visible in the served config
        # (same semantics as the HTTP config route)
        override = self.core.config_overrides.get(model.name, {})
        if "max_batch_size" in override:
            cfg.max_batch_size = int(override["max_batch_size"])
        if "backend" in override:
            cfg.backend = override["backend"]
        if "platform" in override:
            cfg.platform = override["platform"]
        return resp

    def ModelInfer(self, request, context):
        req, raw_list = self._request_to_dict(request)
        response_dict, binary_parts = self.core.infer(
            request.model_name, req, raw_list
        )
        return self._dict_to_response(response_dict, binary_parts, request.id)

    def ModelStreamInfer(self, request_iterator, context):
        """Bi-di stream: errors are reported in-band via error_message so
        the stream survives a failed request (Triton semantics;
        reference ModelStreamInferResponse grpc_service.proto:821-840)."""
        for request in request_iterator:
            try:
                model = self.core.get_model(request.model_name)
                req, raw_list = self._request_to_dict(request)
                want_empty_final = bool(
                    req.get("parameters", {}).get(
                        "triton_enable_empty_final_response", False
                    )
                )
                if model.decoupled:
                    # one response per yielded chunk
                    inputs = {}
                    cursor = 0
                    for inp in req.get("inputs", []):
                        arr, cursor = self.core._input_array(inp, raw_list, cursor)
                        inputs[inp["name"]] = arr
                    stats = self.core.stats[model.name]
                    for chunk in model.execute_decoupled(
                        inputs, req.get("parameters", {})
                    ):
                        rd, parts = self.core._build_response(
                            model, req, chunk, req.get("parameters", {})
                        )
                        resp = self._dict_to_response(rd, parts, request.id)
                        yield service_pb2.ModelStreamInferResponse(
                            infer_response=resp
                        )
                    stats.inference_count += 1
                    stats.execution_count += 1
                    stats.success_count += 1
                    if want_empty_final:
                        final = service_pb2.ModelInferResponse(
                            model_name=request.model_name, id=request.id
                        )
                        final.parameters["triton_final_response"].bool_param = True
                        yield service_pb2.ModelStreamInferResponse(
                            infer_response=final
                        )
                else:
                    response_dict, binary_parts = self.core.infer(
                        request.model_name, req, raw_list
                    )
                    resp = self._dict_to_response(
                        response_dict, binary_parts, request.id
                    )
                    resp.parameters["triton_final_response"].bool_param = True
                    yield service_pb2.ModelStreamInferResponse(infer_response=resp)
            except InferenceError as e:
                yield service_pb2.ModelStreamInferResponse(error_message=str(e))
            except Exception as e:
                yield service_pb2.ModelStreamInferResponse(error_message=str(e))

    def ModelStatistics(self, request, context):
        stats = self.core.statistics(request.name or None)
        resp = service_pb2.ModelStatisticsResponse()
        for ms in stats["model_stats"]:
            m = resp.model_stats.add()
            m.name = ms["name"]
            m.version = ms["version"]
            m.last_inference = ms["last_inference"]
            m.inference_count = ms["inference_count"]
            m.execution_count = ms["execution_count"]
            infer_stats = ms["inference_stats"]
            for key in ("success", "fail", "queue", "compute_input",
                        "compute_infer", "compute_output", "cache_hit",
                        "cache_miss"):
                d = getattr(m.inference_stats, key)
                d.count = infer_stats[key]["count"]
                d.ns = infer_stats[key]["ns"]
        return resp

    def RepositoryIndex(self, request, context):
        resp = service_pb2.RepositoryIndexResponse()
        for entry in self.core.repository_index():
            m = resp.models.add()
            m.name = entry["name"]
            m.version = entry["version"]
            m.state = entry["state"]
            m.reason = entry["reason"]
        return resp

    def RepositoryModelLoad(self, request, context):
        config = None
        files = {}
        for key, param in request.parameters.items():
            if key == "config":
                config = param.string_param
            else:
                files[key] = param.bytes_param
        self.core.load_model(request.model_name, config=config,
                             files=files or None)
        return service_pb2.RepositoryModelLoadResponse()

    def RepositoryModelUnload(self, request, context):
        self.core.unload_model(request.model_name)
        return service_pb2.RepositoryModelUnloadResponse()

    def SystemSharedMemoryStatus(self, request, context):
        resp = service_pb2.SystemSharedMemoryStatusResponse()
        for r in self.core.shm.system_status(request.name or None):
            resp.regions[r["name"]].name = r["name"]
            resp.regions[r["name"]].key = r["key"]
            resp.regions[r["name"]].offset = r["offset"]
            resp.regions[r["name"]].byte_size = r["byte_size"]
        return resp

    def SystemSharedMemoryRegister(self, request, context):
        try:
            self.core.shm.register_system(
                request.name, request.key, request.offset, request.byte_size
            )
        except FileNotFoundError:
            raise InferenceError(
                f"Unable to open shared memory region: '{request.key}'"
            )
        return service_pb2.SystemSharedMemoryRegisterResponse()

    def SystemSharedMemoryUnregister(self, request, context):
        self.core.shm.unregister_system(request.name or None)
        return service_pb2.SystemSharedMemoryUnregisterResponse()

    def CudaSharedMemoryStatus(self, request, context):
        resp = service_pb2.CudaSharedMemoryStatusResponse()
        for r in self.core.shm.hip_status(request.name or None):
            resp.regions[r["name"]].name = r["name"]
            resp.regions[r["name"]].device_id = r["device_id"]
            resp.regions[r["name"]].byte_size = r["byte_size"]
        return resp

    def CudaSharedMemoryRegister(self, request, context):
        self.core.shm.register_hip(
            request.name, request.raw_handle, request.device_id, request.byte_size
        )
        return service_pb2.CudaSharedMemoryRegisterResponse()

    def CudaSharedMemoryUnregister(self, request, context):
        self.core.shm.unregister_hip(request.name or None)
        return service_pb2.CudaSharedMemoryUnregisterResponse()

    def TraceSetting(self, request, context):
        if request.settings:
            for key, val in request.settings.items():
                if len(val.value):
                    self.core.trace_settings[key] = list(val.value)
                else:
                    self.core.trace_settings.pop(key, None)
        resp = service_pb2.TraceSettingResponse()
        for key, val in self.core.trace_settings.items():
            if isinstance(val, (list, tuple)):
                resp.settings[key].value.extend([str(v) for v in val])
            else:
                resp.settings[key].value.append(str(val))
        return resp

    def LogSettings(self, request, context):
        if request.settings:
            for key, val in request.settings.items():
                which = val.WhichOneof("parameter_choice")
                if which is not None:
                    self.core.log_settings[key] = getattr(val, which)
        resp = service_pb2.LogSettingsResponse()
        for key, val in self.core.log_settings.items():
            if isinstance(val, bool):
                resp.settings[key].bool_param = val
            elif isinstance(val, int):
                resp.settings[key].uint32_param = val
            else:
                resp.settings[key].string_param = str(val)
        return resp

    # ---- generic handler ----

    def _make_handler(self):
        server = self

        class Handler(grpc.GenericRpcHandler):
            def service(self, handler_call_details):
                method = handler_call_details.method
                if not method.startswith(f"/{SERVICE_NAME}/"):
                    return None
                rpc_name = method.rsplit("/", 1)[-1]
                if rpc_name not in RPCS:
                    return None
                req_cls, resp_cls, streaming = RPCS[rpc_name]
                impl = getattr(server, rpc_name)

                def _wrap_unary(request, context):
                    try:
                        return impl(request, context)
                    except InferenceError as e:
                        context.abort(grpc.StatusCode.INVALID_ARGUMENT
                                      if e.status != 404
                                      else grpc.StatusCode.NOT_FOUND, str(e))
                    except Exception as e:
                        context.abort(grpc.StatusCode.INTERNAL, str(e))

                if streaming:
                    return grpc.stream_stream_rpc_method_handler(
                        impl,
                        request_deserializer=req_cls.FromString,
                        response_serializer=resp_cls.SerializeToString,
                    )
                return grpc.unary_unary_rpc_method_handler(
                    _wrap_unary,
                    request_deserializer=req_cls.FromString,
                    response_serializer=resp_cls.SerializeToString,
                )

        return Handler()

    # ---- lifecycle ----

    def start(self):
        self._server.start()
        return self

    def stop(self, grace=None):
        self._server.stop(grace)
