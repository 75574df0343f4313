"""KServe-v2 HTTP/1.1 server frontend.

A hand-rolled asyncio.Protocol HTTP server (keep-alive, Content-Length
framing) over InferenceCore. Serves every endpoint the client stack
exercises: health, metadata, config, repository control, infer with
binary tensor framing (``Inference-Header-Content-Length``), statistics,
trace/log settings, and system/cuda(HIP) shared-memory registration
(HTTP route list per reference http_client.cc:1394-1764).
"""

import asyncio
import base64
import gzip
import json
import re
import zlib
from concurrent.futures import ThreadPoolExecutor

from .core import InferenceCore, InferenceError

_ROUTE_INFER = re.compile(r"^/v2/models/([^/]+)(?:/versions/([^/]+))?/infer$")
_ROUTE_READY = re.compile(r"^/v2/models/([^/]+)(?:/versions/([^/]+))?/ready$")
_ROUTE_CONFIG = re.compile(r"^/v2/models/([^/]+)(?:/versions/([^/]+))?/config$")
_ROUTE_STATS = re.compile(r"^/v2/models/([^/]+)(?:/versions/([^/]+))?/stats$")
_ROUTE_META = re.compile(r"^/v2/models/([^/]+)(?:/versions/([^/]+))?$")
_ROUTE_LOAD = re.compile(r"^/v2/repository/models/([^/]+)/load$")
_ROUTE_UNLOAD = re.compile(r"^/v2/repository/models/([^/]+)/unload$")
_ROUTE_SYSSHM = re.compile(
    r"^/v2/systemsharedmemory(?:/region/([^/]+))?/(status|register|unregister)$"
)
_ROUTE_CUDASHM = re.compile(
    r"^/v2/cudasharedmemory(?:/region/([^/]+))?/(status|register|unregister)$"
)
_ROUTE_TRACE = re.compile(r"^/v2(?:/models/([^/]+))?/trace/setting$")


class _HttpProtocol(asyncio.Protocol):
    def __init__(self, server):
        self._server = server
        self._core = server.core
        self._buf = bytearray()
        self._transport = None
        # per-connection FIFO: requests are handled on the executor (GPU
        # models must not block the event loop) but responses must go
        # out in request order on a keep-alive connection
        self._chain = None

    def connection_made(self, transport):
        self._transport = transport
        try:
            import socket

            sock = transport.get_extra_info("socket")
            if sock is not None:
                sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        except Exception:
            pass

    def data_received(self, data):
        self._buf.extend(data)
        while True:
            end = self._buf.find(b"\r\n\r\n")
            if end < 0:
                return
            header_blob = bytes(self._buf[:end])
            lines = header_blob.split(b"\r\n")
            try:
                method, path, _ = lines[0].split(b" ", 2)
            except ValueError:
                self._transport.close()
                return
            headers = {}
            for line in lines[1:]:
                k, _, v = line.partition(b":")
                headers[k.strip().lower().decode("latin1")] = v.strip().decode(
                    "latin1"
                )
            content_length = int(headers.get("content-length", "0"))
            total = end + 4 + content_length
            if len(self._buf) < total:
                return
            body = bytes(self._buf[end + 4 : total])
            del self._buf[:total]
            self._schedule(method.decode(), path.decode(), headers, body)

    def _schedule(self, method, path, headers, body):
        loop = asyncio.get_event_loop()

        async def run(prev):
            reply = await loop.run_in_executor(
                self._server.executor, self._process, method, path, headers,
                body,
            )
            if prev is not None:
                await prev
            if self._transport is not None and not self._transport.is_closing():
                self._transport.write(reply)

        self._chain = loop.create_task(run(self._chain))

    def _process(self, method, path, headers, body):
        """Runs on the executor; returns the raw response bytes."""
        return self._handle(method, path, headers, body)

    def _handle(self, method, path, headers, body):
        enc = headers.get("content-encoding")
        if enc == "gzip":
            body = gzip.decompress(body)
        elif enc == "deflate":
            body = zlib.decompress(body)
        if "?" in path:
            path = path.split("?", 1)[0]
        try:
            status, resp_headers, resp_body = self._server.route(
                method, path, headers, body
            )
        except InferenceError as e:
            resp_body = json.dumps({"error": str(e)}).encode()
            status = e.status
            resp_headers = {"Content-Type": "application/json"}
        except Exception as e:  # pragma: no cover - defensive
            resp_body = json.dumps({"error": str(e)}).encode()
            status = 500
            resp_headers = {"Content-Type": "application/json"}
        # whole-body response compression when the client asked for it
        # (tritonclient response_compression_algorithm sends
        # Accept-Encoding; InferResult gunzips the whole body)
        accept = headers.get("accept-encoding", "")
        if status == 200 and len(resp_body) > 256 \
                and "content-encoding" not in {k.lower()
                                               for k in resp_headers}:
            if "gzip" in accept:
                resp_body = gzip.compress(resp_body)
                resp_headers["Content-Encoding"] = "gzip"
            elif "deflate" in accept:
                resp_body = zlib.compress(resp_body)
                resp_headers["Content-Encoding"] = "deflate"
        reason = {200: "OK", 400: "Bad Request", 404: "Not Found",
                  500: "Internal Server Error"}.get(status, "OK")
        out = [f"HTTP/1.1 {status} {reason}\r\n".encode()]
        resp_headers.setdefault("Content-Type", "application/json")
        resp_headers["Content-Length"] = str(len(resp_body))
        for k, v in resp_headers.items():
            out.append(f"{k}: {v}\r\n".encode())
        out.append(b"\r\n")
        out.append(resp_body)
        return b"".join(out)


class HttpServer:
    """asyncio KServe-v2 HTTP server around an InferenceCore."""

    def __init__(self, core=None, host="127.0.0.1", port=8000, workers=8):
        self.core = core if core is not None else InferenceCore()
        self.host = host
        self.port = port
        self.executor = ThreadPoolExecutor(max_workers=workers)
        self._server = None

    # ---- routing ----
    def route(self, method, path, headers, body):
        core = self.core
        if path == "/v2/health/live":
            return (200 if core.live else 400), {}, b""
        if path == "/v2/health/ready":
            return (200 if core.ready else 400), {}, b""
        if path == "/v2" or path == "/v2/":
            meta = {
                "name": core.server_name,
                "version": core.version,
                "extensions": [
                    "classification", "sequence", "model_repository",
                    "schedule_policy", "model_configuration", "system_shared_memory",
                    "cuda_shared_memory", "binary_tensor_data", "statistics",
                    "trace", "logging",
                ],
            }
            return 200, {}, json.dumps(meta).encode()

        m = _ROUTE_INFER.match(path)
        if m and method == "POST":
            return self._infer(m.group(1), headers, body)
        m = _ROUTE_READY.match(path)
        if m:
            state = core.model_state.get(m.group(1))
            return (200 if state == "READY" else 400), {}, b""
        m = _ROUTE_CONFIG.match(path)
        if m:
            model = core.get_model(m.group(1), must_be_ready=False)
            return 200, {}, json.dumps(core.model_config_dict(model)).encode()
        m = _ROUTE_STATS.match(path)
        if m:
            return 200, {}, json.dumps(core.statistics(m.group(1))).encode()
        if path == "/v2/models/stats":
            return 200, {}, json.dumps(core.statistics()).encode()
        m = _ROUTE_META.match(path)
        if m:
            model = core.get_model(m.group(1), must_be_ready=False)
            return 200, {}, json.dumps(model.metadata()).encode()
        if path == "/v2/repository/index" and method == "POST":
            return 200, {}, json.dumps(core.repository_index()).encode()
        m = _ROUTE_LOAD.match(path)
        if m and method == "POST":
            config = None
            files = None
            if body:
                try:
                    params = json.loads(body.decode()).get("parameters", {})
                except Exception:
                    params = {}
                config = params.get("config")
                files = {
                    k: base64.b64decode(v)
                    for k, v in params.items()
                    if k != "config" and isinstance(v, str)
                } or None
            core.load_model(m.group(1), config=config, files=files)
            return 200, {}, b"{}"
        m = _ROUTE_UNLOAD.match(path)
        if m and method == "POST":
            core.unload_model(m.group(1))
            return 200, {}, b"{}"
        m = _ROUTE_SYSSHM.match(path)
        if m:
            return self._system_shm(m.group(1), m.group(2), method, body)
        m = _ROUTE_CUDASHM.match(path)
        if m:
            return self._cuda_shm(m.group(1), m.group(2), method, body)
        m = _ROUTE_TRACE.match(path)
        if m:
            if method == "POST":
                settings = json.loads(body) if body else {}
                for k, v in settings.items():
                    if v is None:
                        core.trace_settings.pop(k, None)
                    else:
                        core.trace_settings[k] = v
            return 200, {}, json.dumps(core.trace_settings).encode()
        if path == "/v2/logging":
            if method == "POST":
                settings = json.loads(body) if body else {}
                core.log_settings.update(settings)
            return 200, {}, json.dumps(core.log_settings).encode()

        raise InferenceError(f"unknown request path {path}", status=404)

    def _infer(self, model_name, headers, body):
        hlen = headers.get("inference-header-content-length")
        if hlen is not None:
            json_len = int(hlen)
            request = json.loads(body[:json_len])
            binary_buf = body[json_len:]
        else:
            request = json.loads(body) if body else {}
            binary_buf = b""
        response, binary_parts = self.core.infer(model_name, request, binary_buf)
        resp_json = json.dumps(response).encode()
        resp_headers = {}
        # ORCA per-response load metrics (reference README.md:352-366):
        # the client opts in via endpoint-load-metrics-format: text|json.
        fmt = headers.get("endpoint-load-metrics-format")
        model = self.core.models.get(model_name)
        metrics_fn = getattr(model, "load_metrics", None)
        if fmt in ("text", "json") and metrics_fn is not None:
            metrics = metrics_fn()
            if metrics:
                if fmt == "json":
                    resp_headers["endpoint-load-metrics"] = (
                        "JSON " + json.dumps(metrics))
                else:
                    resp_headers["endpoint-load-metrics"] = "TEXT " + ", ".join(
                        f"{k}={v:g}" for k, v in metrics.items())
        if binary_parts:
            resp_headers["Inference-Header-Content-Length"] = str(len(resp_json))
            resp_body = resp_json + b"".join(binary_parts)
        else:
            resp_body = resp_json
        return 200, resp_headers, resp_body

    def _system_shm(self, region, action, method, body):
        core = self.core
        if action == "status":
            return 200, {}, json.dumps(core.shm.system_status(region)).encode()
        if action == "register":
            req = json.loads(body)
            try:
                core.shm.register_system(
                    region, req["key"], req.get("offset", 0), req["byte_size"]
                )
            except FileNotFoundError:
                raise InferenceError(
                    f"Unable to open shared memory region: '{req['key']}'"
                )
            return 200, {}, b"{}"
        if action == "unregister":
            core.shm.unregister_system(region)
            return 200, {}, b"{}"
        raise InferenceError("bad shared memory action", status=400)

    def _cuda_shm(self, region, action, method, body):
        import base64

        core = self.core
        if action == "status":
            return 200, {}, json.dumps(core.shm.hip_status(region)).encode()
        if action == "register":
            req = json.loads(body)
            raw_handle = base64.b64decode(req["raw_handle"]["b64"])
            core.shm.register_hip(
                region, raw_handle, req.get("device_id", 0), req["byte_size"]
            )
            return 200, {}, b"{}"
        if action == "unregister":
            core.shm.unregister_hip(region)
            return 200, {}, b"{}"
        raise InferenceError("bad shared memory action", status=400)

    # ---- lifecycle ----
    async def start(self, ssl_context=None):
        loop = asyncio.get_running_loop()
        self._server = await loop.create_server(
            lambda: _HttpProtocol(self), self.host, self.port,
            ssl=ssl_context,
        )
        if self.port == 0:
            self.port = self._server.sockets[0].getsockname()[1]
        return self

    async def stop(self):
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()

    def serve_forever_in_thread(self, ssl_context=None):
        """Run the server on a dedicated event-loop thread; returns a
        callable that stops it. Used by test fixtures and bench."""
        import threading

        loop = asyncio.new_event_loop()
        started = threading.Event()

        def _run():
            asyncio.set_event_loop(loop)
            loop.run_until_complete(self.start(ssl_context))
            started.set()
            loop.run_forever()

        thread = threading.Thread(target=_run, daemon=True)
        thread.start()
        started.wait(10)

        def _stop():
            async def _shutdown():
                await self.stop()
                loop.stop()

            asyncio.run_coroutine_threadsafe(_shutdown(), loop)
            thread.join(timeout=5)

        return _stop
