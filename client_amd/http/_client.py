"""Synchronous KServe-v2 HTTP/REST client.

API-compatible with ``tritonclient.http.InferenceServerClient``
(reference: tritonclient/http/_client.py). The reference rides on
geventhttpclient + a greenlet pool; this implementation owns its
transport: a keep-alive pool of ``concurrency`` persistent HTTP/1.1
connections plus a thread pool for ``async_infer`` futures. All v2
endpoints are covered, including the log-settings endpoints that exist
only in the Python reference client (reference http/_client.py:867-944).
"""

import base64
import gzip
import http.client
import json
import threading
import urllib.parse
import zlib
from concurrent.futures import ThreadPoolExecutor

from .._client import InferenceServerClientBase
from .._request import Request
from ..utils import InferenceServerException, raise_error
from ._infer_result import InferResult
from ._utils import _get_inference_request, _raise_if_error


class InferAsyncRequest:
    """Handle for an in-flight async_infer; ``get_result()`` blocks until
    the response is available (reference http/_client.py:46-100)."""

    def __init__(self, future, verbose=False):
        self._future = future
        self._verbose = verbose

    def get_result(self, block=True, timeout=None):
        if not block and not self._future.done():
            raise_error("timeout exceeded when not blocking for result")
        try:
            return self._future.result(timeout=timeout)
        except InferenceServerException:
            raise
        except Exception as e:
            raise InferenceServerException(msg=str(e)) from e


class _PooledConnection:
    """One persistent HTTP/1.1 connection with its own lock-free usage
    (pool hands a connection to exactly one request at a time)."""

    def __init__(self, host, port, timeout, ssl_context=None):
        self._host = host
        self._port = port
        self._timeout = timeout
        self._ssl_context = ssl_context
        self._conn = None

    def _connect(self):
        if self._ssl_context is not None:
            self._conn = http.client.HTTPSConnection(
                self._host, self._port, timeout=self._timeout,
                context=self._ssl_context,
            )
        else:
            self._conn = http.client.HTTPConnection(
                self._host, self._port, timeout=self._timeout
            )
        self._conn.connect()
        # Disable Nagle: small JSON requests must not wait for ACKs
        # (the reference sets TCP_NODELAY via curl, http_client.cc:2172-2174).
        import socket

        self._conn.sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)

    def request(self, method, path, body, headers):
        """Issue a request, reconnecting once on a stale keep-alive socket."""
        for attempt in (0, 1):
            if self._conn is None:
                self._connect()
            try:
                self._conn.request(method, path, body=body, headers=headers)
                resp = self._conn.getresponse()
                body_bytes = resp.read()
                return resp, body_bytes
            except (
                http.client.HTTPException,
                ConnectionError,
                BrokenPipeError,
                OSError,
            ):
                self.close()
                if attempt == 1:
                    raise
        raise ConnectionError("unreachable")

    def close(self):
        if self._conn is not None:
            try:
                self._conn.close()
            except Exception:
                pass
            self._conn = None


class _Response:
    """Response facade handed to InferResult: .headers mapping + .read()."""

    __slots__ = ("headers", "_body", "status")

    def __init__(self, status, headers, body):
        self.status = status
        self.headers = headers
        self._body = body

    def read(self):
        return self._body


class InferenceServerClient(InferenceServerClientBase):
    def __init__(
        self,
        url,
        verbose=False,
        concurrency=1,
        connection_timeout=60.0,
        network_timeout=60.0,
        max_greenlets=None,
        ssl=False,
        ssl_options=None,
        ssl_context_factory=None,
        insecure=False,
    ):
        super().__init__()
        if url.startswith("http://") or url.startswith("https://"):
            raise_error("url should not include the scheme")
        parsed = urllib.parse.urlparse("http://" + url)
        self._host = parsed.hostname
        self._port = parsed.port if parsed.port is not None else (
            443 if ssl else 80)
        self._base_path = parsed.path.rstrip("/")
        self._verbose = verbose
        self._concurrency = concurrency
        self._timeout = network_timeout
        ssl_context = None
        if ssl:
            import ssl as ssl_mod

            if ssl_context_factory is not None:
                ssl_context = ssl_context_factory()
            else:
                ssl_context = ssl_mod.create_default_context()
                if insecure:
                    ssl_context.check_hostname = False
                    ssl_context.verify_mode = ssl_mod.CERT_NONE
        self._pool = []
        self._pool_lock = threading.Lock()
        self._pool_sem = threading.Semaphore(concurrency)
        for _ in range(concurrency):
            self._pool.append(
                _PooledConnection(self._host, self._port, network_timeout,
                                  ssl_context)
            )
        self._executor = ThreadPoolExecutor(max_workers=concurrency)
        self._closed = False

    # -- lifecycle -----------------------------------------------------

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
        if self._closed:
            return
        self._closed = True
        self._executor.shutdown(wait=True)
        with self._pool_lock:
            for conn in self._pool:
                conn.close()

    # -- transport -----------------------------------------------------

    def _checkout(self):
        self._pool_sem.acquire()
        with self._pool_lock:
            return self._pool.pop()

    def _checkin(self, conn):
        with self._pool_lock:
            self._pool.append(conn)
        self._pool_sem.release()

    def _validate_headers(self, headers):
        """Reject headers that would corrupt the framed body
        (reference http/_client.py:309-331)."""
        if headers is not None:
            for key in headers.keys():
                if key.lower() == "transfer-encoding":
                    raise_error(
                        "Transfer-Encoding is not allowed as a request header"
                    )

    def _request(self, method, request_uri, body=None, headers=None, query_params=None):
        self._validate_headers(headers)
        hdrs = dict(headers) if headers else {}
        request = Request(hdrs)
        self._call_plugin(request)
        hdrs = request.headers
        path = self._base_path + "/" + request_uri
        if query_params:
            path += "?" + urllib.parse.urlencode(query_params)
        if self._verbose:
            print(f"{method} {path}, headers {hdrs}")
        conn = self._checkout()
        try:
            resp, body_bytes = conn.request(method, path, body, hdrs)
        except OSError as e:
            raise InferenceServerException(msg=str(e)) from e
        finally:
            self._checkin(conn)
        response = _Response(resp.status, resp.headers, body_bytes)
        if self._verbose:
            print(response.status)
        return response

    def _get(self, request_uri, headers=None, query_params=None):
        return self._request("GET", request_uri, None, headers, query_params)

    def _post(self, request_uri, request_body=None, headers=None, query_params=None):
        return self._request("POST", request_uri, request_body, headers, query_params)

    # -- health / metadata --------------------------------------------

    def is_server_live(self, headers=None, query_params=None):
        response = self._get("v2/health/live", headers, query_params)
        return response.status == 200

    def is_server_ready(self, headers=None, query_params=None):
        response = self._get("v2/health/ready", headers, query_params)
        return response.status == 200

    def is_model_ready(self, model_name, model_version="", headers=None, query_params=None):
        if model_version != "":
            request_uri = "v2/models/{}/versions/{}/ready".format(
                urllib.parse.quote(model_name), model_version
            )
        else:
            request_uri = "v2/models/{}/ready".format(urllib.parse.quote(model_name))
        response = self._get(request_uri, headers, query_params)
        return response.status == 200

    def get_server_metadata(self, headers=None, query_params=None):
        response = self._get("v2", headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    def get_model_metadata(
        self, model_name, model_version="", headers=None, query_params=None
    ):
        if model_version != "":
            request_uri = "v2/models/{}/versions/{}".format(
                urllib.parse.quote(model_name), model_version
            )
        else:
            request_uri = "v2/models/{}".format(urllib.parse.quote(model_name))
        response = self._get(request_uri, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    def get_model_config(
        self, model_name, model_version="", headers=None, query_params=None
    ):
        if model_version != "":
            request_uri = "v2/models/{}/versions/{}/config".format(
                urllib.parse.quote(model_name), model_version
            )
        else:
            request_uri = "v2/models/{}/config".format(urllib.parse.quote(model_name))
        response = self._get(request_uri, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    # -- model repository ---------------------------------------------

    def get_model_repository_index(self, headers=None, query_params=None):
        response = self._post("v2/repository/index", None, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    def load_model(
        self, model_name, headers=None, query_params=None, config=None, files=None
    ):
        """Load/reload a model; optional config override and base64 file
        overrides (reference http/_client.py: load + config/file override;
        C++ http_client.cc:1504-1547)."""
        request_uri = "v2/repository/models/{}/load".format(
            urllib.parse.quote(model_name)
        )
        load_request = {}
        if config is not None:
            if "parameters" not in load_request:
                load_request["parameters"] = {}
            load_request["parameters"]["config"] = config
        if files is not None:
            for path, content in files.items():
                if "parameters" not in load_request:
                    load_request["parameters"] = {}
                load_request["parameters"][path] = base64.b64encode(content).decode(
                    "ascii"
                )
        response = self._post(
            request_uri, json.dumps(load_request), headers, query_params
        )
        _raise_if_error(response.status, response.read())
        if self._verbose:
            print("Loaded model '{}'".format(model_name))

    def unload_model(
        self, model_name, headers=None, query_params=None, unload_dependents=False
    ):
        request_uri = "v2/repository/models/{}/unload".format(
            urllib.parse.quote(model_name)
        )
        unload_request = {
            "parameters": {"unload_dependents": unload_dependents}
        }
        response = self._post(
            request_uri, json.dumps(unload_request), headers, query_params
        )
        _raise_if_error(response.status, response.read())
        if self._verbose:
            print("Unloaded model '{}'".format(model_name))

    # -- statistics / trace / logging ---------------------------------

    def get_inference_statistics(
        self, model_name="", model_version="", headers=None, query_params=None
    ):
        if model_name != "":
            if model_version != "":
                request_uri = "v2/models/{}/versions/{}/stats".format(
                    urllib.parse.quote(model_name), model_version
                )
            else:
                request_uri = "v2/models/{}/stats".format(
                    urllib.parse.quote(model_name)
                )
        else:
            request_uri = "v2/models/stats"
        response = self._get(request_uri, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    def update_trace_settings(
        self, model_name=None, settings={}, headers=None, query_params=None
    ):
        if model_name is not None and model_name != "":
            request_uri = "v2/models/{}/trace/setting".format(
                urllib.parse.quote(model_name)
            )
        else:
            request_uri = "v2/trace/setting"
        response = self._post(request_uri, json.dumps(settings), headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    def get_trace_settings(self, model_name=None, headers=None, query_params=None):
        if model_name is not None and model_name != "":
            request_uri = "v2/models/{}/trace/setting".format(
                urllib.parse.quote(model_name)
            )
        else:
            request_uri = "v2/trace/setting"
        response = self._get(request_uri, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    def update_log_settings(self, settings, headers=None, query_params=None):
        """Python-only endpoint in the reference (http/_client.py:867-905)."""
        request_uri = "v2/logging"
        response = self._post(request_uri, json.dumps(settings), headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    def get_log_settings(self, headers=None, query_params=None):
        request_uri = "v2/logging"
        response = self._get(request_uri, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    # -- shared memory management -------------------------------------

    def get_system_shared_memory_status(
        self, region_name="", headers=None, query_params=None
    ):
        if region_name != "":
            request_uri = "v2/systemsharedmemory/region/{}/status".format(
                urllib.parse.quote(region_name)
            )
        else:
            request_uri = "v2/systemsharedmemory/status"
        response = self._get(request_uri, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    def register_system_shared_memory(
        self, name, key, byte_size, offset=0, headers=None, query_params=None
    ):
        request_uri = "v2/systemsharedmemory/region/{}/register".format(
            urllib.parse.quote(name)
        )
        register_request = {"key": key, "offset": offset, "byte_size": byte_size}
        response = self._post(
            request_uri, json.dumps(register_request), headers, query_params
        )
        _raise_if_error(response.status, response.read())
        if self._verbose:
            print("Registered system shared memory with name '{}'".format(name))

    def unregister_system_shared_memory(
        self, name="", headers=None, query_params=None
    ):
        if name != "":
            request_uri = "v2/systemsharedmemory/region/{}/unregister".format(
                urllib.parse.quote(name)
            )
        else:
            request_uri = "v2/systemsharedmemory/unregister"
        response = self._post(request_uri, None, headers, query_params)
        _raise_if_error(response.status, response.read())
        if self._verbose:
            if name != "":
                print("Unregistered system shared memory with name '{}'".format(name))
            else:
                print("Unregistered all system shared memory regions")

    def get_cuda_shared_memory_status(
        self, region_name="", headers=None, query_params=None
    ):
        """HIP-IPC regions are registered under the cuda endpoints so the
        wire protocol stays byte-compatible with a ROCm Triton server
        (reference http_client.cc:1708-1748; the handle is 64 bytes in
        both runtimes)."""
        if region_name != "":
            request_uri = "v2/cudasharedmemory/region/{}/status".format(
                urllib.parse.quote(region_name)
            )
        else:
            request_uri = "v2/cudasharedmemory/status"
        response = self._get(request_uri, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    def register_cuda_shared_memory(
        self, name, raw_handle, device_id, byte_size, headers=None, query_params=None
    ):
        """``raw_handle`` is the base64-encoded hipIpcMemHandle_t bytes
        (reference http/_client.py register_cuda_shared_memory)."""
        request_uri = "v2/cudasharedmemory/region/{}/register".format(
            urllib.parse.quote(name)
        )
        if isinstance(raw_handle, bytes):
            b64_handle = raw_handle.decode("ascii")
        else:
            b64_handle = raw_handle
        register_request = {
            "raw_handle": {"b64": b64_handle},
            "device_id": device_id,
            "byte_size": byte_size,
        }
        response = self._post(
            request_uri, json.dumps(register_request), headers, query_params
        )
        _raise_if_error(response.status, response.read())
        if self._verbose:
            print("Registered cuda shared memory with name '{}'".format(name))

    # AMD-native spelling; same endpoint.
    register_hip_shared_memory = register_cuda_shared_memory
    get_hip_shared_memory_status = get_cuda_shared_memory_status

    def unregister_cuda_shared_memory(self, name="", headers=None, query_params=None):
        if name != "":
            request_uri = "v2/cudasharedmemory/region/{}/unregister".format(
                urllib.parse.quote(name)
            )
        else:
            request_uri = "v2/cudasharedmemory/unregister"
        response = self._post(request_uri, None, headers, query_params)
        _raise_if_error(response.status, response.read())
        if self._verbose:
            if name != "":
                print("Unregistered cuda shared memory with name '{}'".format(name))
            else:
                print("Unregistered all cuda shared memory regions")

    unregister_hip_shared_memory = unregister_cuda_shared_memory

    # -- inference -----------------------------------------------------

    @staticmethod
    def generate_request_body(
        inputs,
        request_id="",
        sequence_id=0,
        sequence_start=False,
        sequence_end=False,
        priority=0,
        timeout=None,
        outputs=None,
        parameters=None,
    ):
        """Stateless body builder for out-of-band use (reference
        http/_client.py:1219-1270). Returns (body, json_size|None)."""
        return _get_inference_request(
            inputs=inputs,
            request_id=request_id,
            outputs=outputs,
            sequence_id=sequence_id,
            sequence_start=sequence_start,
            sequence_end=sequence_end,
            priority=priority,
            timeout=timeout,
            custom_parameters=parameters,
        )

    @staticmethod
    def parse_response_body(
        response_body, verbose=False, header_length=None, content_encoding=None
    ):
        """Stateless response parser (reference http/_client.py:1290-1330)."""
        return InferResult.from_response_body(
            response_body, verbose, header_length, content_encoding
        )

    def _build_infer(
        self,
        model_name,
        inputs,
        model_version,
        outputs,
        request_id,
        sequence_id,
        sequence_start,
        sequence_end,
        priority,
        timeout,
        request_compression_algorithm,
        parameters,
    ):
        request_body, json_size = _get_inference_request(
            inputs=inputs,
            request_id=request_id,
            outputs=outputs,
            sequence_id=sequence_id,
            sequence_start=sequence_start,
            sequence_end=sequence_end,
            priority=priority,
            timeout=timeout,
            custom_parameters=parameters,
        )
        headers = {}
        if request_compression_algorithm == "gzip":
            headers["Content-Encoding"] = "gzip"
            request_body = gzip.compress(request_body)
        elif request_compression_algorithm == "deflate":
            headers["Content-Encoding"] = "deflate"
            request_body = zlib.compress(request_body)
        if json_size is not None:
            headers["Inference-Header-Content-Length"] = str(json_size)
        if model_version != "":
            request_uri = "v2/models/{}/versions/{}/infer".format(
                urllib.parse.quote(model_name), model_version
            )
        else:
            request_uri = "v2/models/{}/infer".format(urllib.parse.quote(model_name))
        return request_uri, request_body, headers

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
        headers=None,
        query_params=None,
        request_compression_algorithm=None,
        response_compression_algorithm=None,
        parameters=None,
    ):
        """Synchronous inference (reference http/_client.py:1331-1481)."""
        request_uri, request_body, extra_headers = self._build_infer(
            model_name,
            inputs,
            model_version,
            outputs,
            request_id,
            sequence_id,
            sequence_start,
            sequence_end,
            priority,
            timeout,
            request_compression_algorithm,
            parameters,
        )
        hdrs = dict(headers) if headers else {}
        hdrs.update(extra_headers)
        if response_compression_algorithm in ("gzip", "deflate"):
            hdrs["Accept-Encoding"] = response_compression_algorithm
        response = self._post(request_uri, request_body, hdrs, query_params)
        _raise_if_error(response.status, response.read())
        return InferResult(response, self._verbose)

    def async_infer(
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
        headers=None,
        query_params=None,
        request_compression_algorithm=None,
        response_compression_algorithm=None,
        parameters=None,
    ):
        """Submit inference on the thread pool; returns InferAsyncRequest.

        The reference returns a greenlet-backed handle
        (http/_client.py:1641-1651); semantics (get_result blocks) match.
        """
        future = self._executor.submit(
            self.infer,
            model_name,
            inputs,
            model_version,
            outputs,
            request_id,
            sequence_id,
            sequence_start,
            sequence_end,
            priority,
            timeout,
            headers,
            query_params,
            request_compression_algorithm,
            response_compression_algorithm,
            parameters,
        )
        return InferAsyncRequest(future, self._verbose)
