"""Protocol-agnostic inference core for the client_amd server.

Shared by the HTTP and gRPC frontends: model repository, shared-memory
registry (system POSIX shm + HIP-IPC device regions), request execution
with per-model statistics in the KServe-v2 stats schema
(reference schema: grpc_service.proto:881-1235).
"""

import threading
import time

import numpy as np

from ..utils import (
    deserialize_bytes_tensor,
    deserialize_bf16_tensor,
    serialize_byte_tensor,
    serialize_bf16_tensor,
    triton_to_np_dtype,
    np_to_triton_dtype,
)


class ShmRegistry:
    """Server-side registry of client-shared memory regions.

    System regions are mapped via POSIX shm_open/mmap; HIP regions are
    opened from the 64-byte hipIpcMemHandle_t with hipIpcOpenMemHandle
    (via client_amd.ops) and stay resident in HBM — tensors never cross
    PCIe.
    """

    def __init__(self):
        self._system = {}
        self._hip = {}
        self._lock = threading.Lock()

    # ---- system shm ----
    def register_system(self, name, key, offset, byte_size):
        import mmap
        import os

        fd = os.open("/dev/shm/" + key.lstrip("/"), os.O_RDWR)
        try:
            mem = mmap.mmap(fd, byte_size + offset)
        finally:
            os.close(fd)
        with self._lock:
            if name in self._system:
                self._system.pop(name)
            self._system[name] = {
                "name": name,
                "key": key,
                "offset": offset,
                "byte_size": byte_size,
                "mmap": mem,
            }

    def unregister_system(self, name=None):
        with self._lock:
            names = [name] if name else list(self._system)
            for n in names:
                region = self._system.pop(n, None)
                if region is not None:
                    region["mmap"].close()

    def system_status(self, name=None):
        with self._lock:
            regions = (
                [self._system[name]] if name and name in self._system
                else list(self._system.values()) if not name else []
            )
            return [
                {"name": r["name"], "key": r["key"], "offset": r["offset"],
                 "byte_size": r["byte_size"]}
                for r in regions
            ]

    # ---- HIP shm ----
    def register_hip(self, name, raw_handle, device_id, byte_size):
        """raw_handle: 64 bytes of hipIpcMemHandle_t."""
        from ..ops import hip_runtime as hr

        ptr = hr.ipc_open_mem_handle(raw_handle)
        with self._lock:
            if name in self._hip:
                old = self._hip.pop(name)
                hr.ipc_close_mem_handle(old["ptr"])
            self._hip[name] = {
                "name": name,
                "device_id": device_id,
                "byte_size": byte_size,
                "ptr": ptr,
            }

    def unregister_hip(self, name=None):
        from ..ops import hip_runtime as hr

        with self._lock:
            names = [name] if name else list(self._hip)
            for n in names:
                region = self._hip.pop(n, None)
                if region is not None:
                    hr.ipc_close_mem_handle(region["ptr"])

    def hip_status(self, name=None):
        with self._lock:
            regions = (
                [self._hip[name]] if name and name in self._hip
                else list(self._hip.values()) if not name else []
            )
            return [
                {"name": r["name"], "device_id": r["device_id"],
                 "byte_size": r["byte_size"]}
                for r in regions
            ]

    def get_system(self, name):
        with self._lock:
            return self._system.get(name)

    def get_hip(self, name):
        with self._lock:
            return self._hip.get(name)


class ModelStats:
    def __init__(self):
        self.inference_count = 0
        self.execution_count = 0
        self.last_inference = 0
        self.success_count = 0
        self.success_ns = 0
        self.fail_count = 0
        self.fail_ns = 0
        self.queue_ns = 0
        self.compute_input_ns = 0
        self.compute_infer_ns = 0
        self.compute_output_ns = 0

    def to_dict(self, name, version="1"):
        def d(count, ns):
            return {"count": count, "ns": ns}

        return {
            "name": name,
            "version": version,
            "last_inference": self.last_inference,
            "inference_count": self.inference_count,
            "execution_count": self.execution_count,
            "inference_stats": {
                "success": d(self.success_count, self.success_ns),
                "fail": d(self.fail_count, self.fail_ns),
                "queue": d(self.success_count, self.queue_ns),
                "compute_input": d(self.success_count, self.compute_input_ns),
                "compute_infer": d(self.success_count, self.compute_infer_ns),
                "compute_output": d(self.success_count, self.compute_output_ns),
                "cache_hit": d(0, 0),
                "cache_miss": d(0, 0),
            },
            "batch_stats": [],
            "memory_usage": [],
        }


class InferenceError(Exception):
    def __init__(self, msg, status=400):
        super().__init__(msg)
        self.status = status


def _check_shm_bounds(region, offset, byte_size):
    """Reject client-supplied (offset, byte_size) windows that fall
    outside the registered region before ANY copy touches the mapping —
    the offsets come off the wire and otherwise index arbitrary server
    memory (HIP regions share the address space with model weights).
    Mirrors the reference server's rejection of out-of-range shm access.
    """
    if (offset < 0 or byte_size < 0
            or offset + byte_size > region["byte_size"]):
        raise InferenceError(
            "Invalid offset + byte size for shared memory region: "
            f"'{region['name']}' (offset {offset} + byte size {byte_size} "
            f"exceeds registered size {region['byte_size']})"
        )


class InferenceCore:
    """Executes KServe-v2 inference requests against the model repository."""

    def __init__(self, server_name="client_amd_server", version="0.1.0"):
        self.server_name = server_name
        self.version = version
        self.models = {}
        self.model_state = {}
        self.stats = {}
        self.shm = ShmRegistry()
        self.trace_settings = {
            "trace_file": "",
            "trace_level": ["OFF"],
            "trace_rate": "1000",
            "trace_count": "-1",
            "log_frequency": "0",
        }
        self.log_settings = {
            "log_file": "",
            "log_info": True,
            "log_warning": True,
            "log_error": True,
            "log_verbose_level": 0,
            "log_format": "default",
        }
        self.live = True
        self.ready = True
        self.config_overrides = {}
        self.file_overrides = {}
        self._tls = threading.local()

    # ---- repository ----
    def add_model(self, model, ready=True):
        self.models[model.name] = model
        self.model_state[model.name] = "READY" if ready else "UNAVAILABLE"
        self.stats.setdefault(model.name, ModelStats())

    def get_model(self, name, must_be_ready=True):
        model = self.models.get(name)
        if model is None:
            raise InferenceError(f"Request for unknown model: '{name}' is not found",
                                 status=404)
        if must_be_ready and self.model_state.get(name) != "READY":
            raise InferenceError(
                f"Request for unknown model: '{name}' is not ready", status=400
            )
        return model

    def load_model(self, name, config=None, files=None):
        """Load/reload; optional config override (JSON string) and file
        overrides ({path: bytes}). Mirrors the reference server's rules:
        a file override requires a config override, and the override is
        visible in the served model config until the next plain load
        (reference http_client.cc:1504-1547 client side)."""
        if name not in self.models:
            raise InferenceError(f"failed to load '{name}', no such model", status=400)
        if files and not config:
            raise InferenceError(
                "File override requires model configuration override"
            )
        if config is not None:
            import json as _json

            try:
                override = _json.loads(config) if isinstance(config, str) \
                    else dict(config)
            except Exception:
                raise InferenceError(
                    f"failed to load '{name}': invalid config override"
                )
            self.config_overrides[name] = override
        else:
            self.config_overrides.pop(name, None)
        self.file_overrides[name] = (
            {path: len(content) for path, content in files.items()}
            if files else {}
        )
        self.model_state[name] = "READY"

    def model_config_dict(self, model):
        """The served config: the model's own config shallow-merged with
        any load-time override."""
        cfg = model.config()
        cfg.update(self.config_overrides.get(model.name, {}))
        return cfg

    def unload_model(self, name):
        if name not in self.models:
            raise InferenceError(f"failed to unload '{name}', no such model",
                                 status=400)
        self.model_state[name] = "UNAVAILABLE"

    def repository_index(self):
        return [
            {"name": name, "version": "1", "state": self.model_state[name],
             "reason": ""}
            for name in self.models
        ]

    def statistics(self, model_name=None):
        if model_name:
            self.get_model(model_name, must_be_ready=False)
            names = [model_name]
        else:
            names = list(self.models)
        return {"model_stats": [self.stats[n].to_dict(n) for n in names]}

    # ---- input validation ----
    @staticmethod
    def _validate_inputs(model, req_inputs):
        """Reject unknown input names and shapes incompatible with the
        model spec (-1 dims are wildcards; one extra leading batch dim
        is allowed — standard batching semantics). The reference server
        rejects both; a client sending a wrong-shape tensor must get an
        error, not a silently reshaped result."""
        lax = getattr(model, "lax_shapes", False)
        spec = {n: list(dims) for n, _, dims in model.inputs}
        for inp in req_inputs:
            name = inp.get("name")
            dims = spec.get(name)
            if dims is None:
                raise InferenceError(
                    f"unexpected inference input '{name}' for model "
                    f"'{model.name}'"
                )
            if lax:
                continue
            shape = list(inp.get("shape") or [])
            cand = shape
            if len(cand) == len(dims) + 1:
                cand = cand[1:]  # leading batch dim
            if len(cand) != len(dims) or any(
                d != -1 and d != s for d, s in zip(dims, cand)
            ):
                raise InferenceError(
                    f"unexpected shape for input '{name}' for model "
                    f"'{model.name}'. Expected {dims}, got {shape}"
                )

    # ---- input materialization ----
    def _input_array(self, inp, binary_buf, binary_cursor):
        """Returns (numpy array, new_cursor). inp is the request-JSON dict."""
        name = inp["name"]
        datatype = inp["datatype"]
        shape = inp["shape"]
        params = inp.get("parameters", {})
        elem_count = int(np.prod(shape)) if shape else 1

        shm_name = params.get("shared_memory_region")
        if shm_name is not None:
            byte_size = params["shared_memory_byte_size"]
            offset = params.get("shared_memory_offset", 0)
            region = self.shm.get_system(shm_name)
            if region is not None:
                _check_shm_bounds(region, offset, byte_size)
                base = region["offset"] + offset
                raw = bytes(region["mmap"][base : base + byte_size])
                return self._decode_raw(raw, datatype, shape), binary_cursor
            hip_region = self.shm.get_hip(shm_name)
            if hip_region is not None:
                from ..ops import hip_runtime as hr

                _check_shm_bounds(hip_region, offset, byte_size)
                raw = hr.memcpy_d2h(
                    hip_region["ptr"] + offset, byte_size, hip_region["device_id"]
                )
                return self._decode_raw(raw, datatype, shape), binary_cursor
            raise InferenceError(
                f"Unable to find shared memory region: '{shm_name}'"
            )

        bsize = params.get("binary_data_size")
        if bsize is not None:
            if isinstance(binary_buf, (list, tuple)):
                # gRPC path: one buffer per input (raw_input_contents)
                raw = binary_buf[binary_cursor]
                cursor = binary_cursor + 1
            else:
                raw = binary_buf[binary_cursor : binary_cursor + bsize]
                cursor = binary_cursor + bsize
            if len(raw) != bsize:
                raise InferenceError(
                    f"expected {bsize} bytes of binary data for input '{name}'"
                )
            return self._decode_raw(raw, datatype, shape), cursor

        data = inp.get("data")
        if data is None:
            raise InferenceError(f"no data supplied for input '{name}'")
        if datatype == "BYTES":
            arr = np.array(
                [v.encode("utf-8") if isinstance(v, str) else v for v in data],
                dtype=np.object_,
            ).reshape(shape)
        elif datatype == "BF16":
            raise InferenceError("BF16 data must be sent as binary")
        else:
            arr = np.array(data, dtype=triton_to_np_dtype(datatype)).reshape(shape)
        return arr, binary_cursor

    @staticmethod
    def _decode_raw(raw, datatype, shape):
        if datatype == "BYTES":
            arr = deserialize_bytes_tensor(raw)
        elif datatype == "BF16":
            arr = deserialize_bf16_tensor(raw)
        else:
            arr = np.frombuffer(raw, dtype=triton_to_np_dtype(datatype)).copy()
        return arr.reshape(shape)

    @staticmethod
    def _encode_raw(arr, datatype):
        if datatype == "BYTES":
            s = serialize_byte_tensor(arr)
            return s.item() if s.size > 0 else b""
        if datatype == "BF16":
            s = serialize_bf16_tensor(arr)
            return s.item() if s.size > 0 else b""
        return np.ascontiguousarray(arr).tobytes()

    # ---- inference ----
    def infer(self, model_name, request, binary_buf=b""):
        """Execute one request.

        request: the parsed KServe-v2 JSON dict; binary_buf: the raw bytes
        that followed the JSON header. Returns (response_dict, binary_parts)
        where binary_parts is a list of raw output buffers to append.
        """
        model = self.get_model(model_name)
        stats = self.stats[model_name]
        t0 = time.monotonic_ns()
        parameters = dict(request.get("parameters", {}))

        try:
            self._validate_inputs(model, request.get("inputs", []))
            device_result = self._try_device_infer(model, request, parameters)
            if device_result is not None:
                response, binary_parts, dt_input, dt_infer = device_result
                t1 = t0 + dt_input
                t2 = t1 + dt_infer
            else:
                inputs = {}
                cursor = 0
                for inp in request.get("inputs", []):
                    arr, cursor = self._input_array(inp, binary_buf, cursor)
                    inputs[inp["name"]] = arr
                t1 = time.monotonic_ns()

                result = model.execute(inputs, parameters)
                t2 = time.monotonic_ns()

                response, binary_parts = self._build_response(
                    model, request, result, parameters
                )
            t3 = time.monotonic_ns()

            stats.inference_count += 1
            stats.execution_count += 1
            stats.success_count += 1
            stats.last_inference = int(time.time() * 1000)
            stats.compute_input_ns += t1 - t0
            stats.compute_infer_ns += t2 - t1
            stats.compute_output_ns += t3 - t2
            stats.success_ns += t3 - t0
            return response, binary_parts
        except InferenceError:
            stats.fail_count += 1
            raise
        except Exception as e:
            stats.fail_count += 1
            raise InferenceError(str(e)) from e

    def _try_device_infer(self, model, request, parameters):
        """Zero-copy HIP-shm fast path: when every input
        and every requested output lives in a registered HIP-IPC region
        and the model executes on the GPU, wrap the regions as torch
        tensors via DLPack (kDLROCM) and never touch the host. Returns
        (response, binary_parts, t_input_ns, t_infer_ns) or None to fall
        back to the generic path."""
        if not hasattr(model, "execute_torch"):
            return None
        req_inputs = request.get("inputs", [])
        req_outputs = request.get("outputs")
        if not req_inputs or not req_outputs:
            return None
        for io in req_inputs + req_outputs:
            params = io.get("parameters", {})
            region_name = params.get("shared_memory_region")
            if region_name is None or self.shm.get_hip(region_name) is None:
                return None

        import torch

        from ..utils._dlpack import DLDeviceType
        from ..utils._shared_memory_tensor import SharedMemoryTensor

        def region_tensor(io, datatype, shape):
            params = io["parameters"]
            region = self.shm.get_hip(params["shared_memory_region"])
            offset = params.get("shared_memory_offset", 0)
            # bound BOTH the declared window and the actual tensor extent
            # the DLPack view will touch
            itemsize = 2 if datatype == "BF16" else np.dtype(
                triton_to_np_dtype(datatype)
            ).itemsize  # BF16 maps to fp32 in the numpy table; wire is 2B
            nbytes = int(np.prod(shape)) * itemsize
            _check_shm_bounds(
                region, offset,
                max(nbytes, params.get("shared_memory_byte_size", 0)),
            )
            smt = SharedMemoryTensor(
                datatype=datatype,
                shape=shape,
                base_addr=region["ptr"] + offset,
                byte_offset=0,
                device_type=DLDeviceType.kDLROCM,
                device_id=region["device_id"],
            )
            return torch.from_dlpack(smt)

        t0 = time.monotonic_ns()
        tensors = []
        for io in req_inputs:
            tensors.append(region_tensor(io, io["datatype"], io["shape"]))
        t1 = time.monotonic_ns()
        if hasattr(model, "execute_torch_async"):
            results, done_ev = model.execute_torch_async(tensors)
        else:
            results, done_ev = model.execute_torch(tensors), None
        t2 = time.monotonic_ns()

        model_dtypes = {n: d for n, d, _ in model.outputs}
        response = {
            "model_name": model.name,
            "model_version": "1",
            "outputs": [],
        }
        if "id" in request:
            response["id"] = request["id"]
        torch_dt = {
            "FP32": torch.float32, "BF16": torch.bfloat16, "FP16": torch.float16,
            "INT64": torch.int64, "INT32": torch.int32, "INT8": torch.int8,
            "UINT8": torch.uint8, "FP64": torch.float64, "BOOL": torch.bool,
        }
        by_name = {n: r for (n, _, _), r in zip(model.outputs, results)}
        # Everything stays on the default stream (a dedicated-copy-
        # stream design measured SLOWER on MI355X — see
        # DynamicBatcher._run). The request host-waits a per-request
        # event recorded after ITS copies instead of a full device
        # synchronize, so it never blocks on other models' side-stream
        # work (e.g. the LLM decode loop).
        for io in req_outputs:
            name = io["name"]
            if name not in by_name:
                raise InferenceError(
                    f"unexpected inference output '{name}' for model "
                    f"'{model.name}'"
                )
            result = by_name[name]
            datatype = model_dtypes[name]
            out_view = region_tensor(
                {"parameters": io["parameters"]}, datatype,
                list(result.shape)
            )
            out_view.copy_(result.to(torch_dt[datatype]))
            params = io["parameters"]
            response["outputs"].append({
                "name": name,
                "datatype": datatype,
                "shape": list(result.shape),
                "parameters": {
                    "shared_memory_region":
                        params["shared_memory_region"],
                    "shared_memory_byte_size":
                        params["shared_memory_byte_size"],
                    **({"shared_memory_offset":
                        params["shared_memory_offset"]}
                       if params.get("shared_memory_offset") else {}),
                },
            })
        self._sync_outputs(torch)
        return response, [], t1 - t0, t2 - t1

    _SYNC_MODE = None

    def _sync_outputs(self, torch):
        """Wait for this request's output copies. Full device
        synchronize by DEFAULT: host event-synchronize measured
        catastrophically slower on MI355X under thread concurrency
        (DenseNet c8: 1928 inf/s with per-request hipEventSynchronize —
        even with reused events — vs 4419 with hipDeviceSynchronize;
        gpurun_out/r02v6). CLIENT_AMD_SYNC_MODE=event keeps the event
        path selectable for re-measurement on future ROCm builds."""
        if InferenceCore._SYNC_MODE is None:
            import os

            InferenceCore._SYNC_MODE = os.environ.get(
                "CLIENT_AMD_SYNC_MODE", "device")
        if InferenceCore._SYNC_MODE == "device":
            torch.cuda.synchronize()
            return
        ev = getattr(self._tls, "sync_event", None)
        if ev is None:
            ev = torch.cuda.Event()
            self._tls.sync_event = ev
        ev.record()
        ev.synchronize()

    def _build_response(self, model, request, result, parameters):
        requested = request.get("outputs")
        binary_default = bool(
            parameters.get("binary_data_output", False)
        )
        out_specs = []
        if requested:
            for out in requested:
                out_specs.append((out["name"], out.get("parameters", {})))
        else:
            for name in result:
                out_specs.append((name, {"binary_data": binary_default}))

        model_dtypes = {n: d for n, d, _ in model.outputs}
        response = {
            "model_name": model.name,
            "model_version": "1",
            "outputs": [],
        }
        if "id" in request:
            response["id"] = request["id"]
        binary_parts = []
        for name, params in out_specs:
            if name not in result:
                raise InferenceError(
                    f"unexpected inference output '{name}' for model '{model.name}'"
                )
            arr = result[name]
            datatype = model_dtypes.get(name) or np_to_triton_dtype(arr.dtype)
            out_json = {
                "name": name,
                "datatype": datatype,
                "shape": list(arr.shape),
            }

            shm_name = params.get("shared_memory_region")
            class_count = params.get("classification", 0)
            if class_count:
                arr = self._classify(arr, class_count)
                out_json["datatype"] = "BYTES"
                out_json["shape"] = list(arr.shape)
                datatype = "BYTES"

            if shm_name is not None:
                raw = self._encode_raw(arr, datatype)
                byte_size = params["shared_memory_byte_size"]
                offset = params.get("shared_memory_offset", 0)
                if len(raw) > byte_size:
                    raise InferenceError(
                        f"shared memory region '{shm_name}' is too small for "
                        f"output '{name}'"
                    )
                region = self.shm.get_system(shm_name)
                if region is not None:
                    _check_shm_bounds(region, offset, len(raw))
                    _check_shm_bounds(region, offset, byte_size)
                    base = region["offset"] + offset
                    region["mmap"][base : base + len(raw)] = raw
                else:
                    hip_region = self.shm.get_hip(shm_name)
                    if hip_region is None:
                        raise InferenceError(
                            f"Unable to find shared memory region: '{shm_name}'"
                        )
                    from ..ops import hip_runtime as hr

                    _check_shm_bounds(hip_region, offset, len(raw))
                    _check_shm_bounds(hip_region, offset, byte_size)
                    hr.memcpy_h2d(
                        hip_region["ptr"] + offset, raw, len(raw),
                        hip_region["device_id"],
                    )
                out_json["parameters"] = {
                    "shared_memory_region": shm_name,
                    "shared_memory_byte_size": len(raw),
                }
                if offset:
                    out_json["parameters"]["shared_memory_offset"] = offset
            elif params.get("binary_data", binary_default):
                raw = self._encode_raw(arr, datatype)
                out_json["parameters"] = {"binary_data_size": len(raw)}
                binary_parts.append(raw)
            else:
                if datatype == "BYTES":
                    out_json["data"] = [
                        v.decode("utf-8") if isinstance(v, bytes) else str(v)
                        for v in arr.reshape(-1)
                    ]
                elif datatype == "BF16":
                    raise InferenceError("BF16 outputs require binary_data")
                else:
                    out_json["data"] = [v.item() for v in arr.reshape(-1)]
            response["outputs"].append(out_json)
        return response, binary_parts

    @staticmethod
    def _classify(arr, k):
        """Top-k classification extension: returns BYTES '<score>:<idx>'."""
        flat = arr.reshape(arr.shape[0], -1) if arr.ndim > 1 else arr.reshape(1, -1)
        rows = []
        for row in flat:
            idx = np.argsort(row)[::-1][:k]
            rows.append([f"{row[i]}:{i}".encode("utf-8") for i in idx])
        out = np.array(rows, dtype=np.object_)
        return out
