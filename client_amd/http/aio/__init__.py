"""client_amd.http.aio — asyncio KServe-v2 HTTP client on aiohttp.

Mirrors tritonclient.http.aio (reference:
tritonclient/http/aio/__init__.py — aiohttp TCPConnector + ClientSession,
all endpoints as coroutines).
"""

import base64
import gzip
import json
import urllib.parse
import zlib

import aiohttp

from ..._client import InferenceServerClientBase
from ..._request import Request
from ...utils import InferenceServerException, raise_error
from .._infer_result import InferResult
from .._infer_input import InferInput  # re-export for API parity
from .._requested_output import InferRequestedOutput  # re-export
from .._utils import _get_inference_request, _raise_if_error

__all__ = [
    "InferenceServerClient",
    "InferInput",
    "InferRequestedOutput",
    "InferResult",
    "InferenceServerException",
]


class _Response:
    __slots__ = ("headers", "_body", "status")

    def __init__(self, status, headers, body):
        self.status = status
        self.headers = headers
        self._body = body

    def read(self):
        return self._body


class InferenceServerClient(InferenceServerClientBase):
    def __init__(self, url, verbose=False, conn_limit=100, conn_timeout=60.0,
                 ssl=False, ssl_context=None):
        super().__init__()
        if url.startswith("http://") or url.startswith("https://"):
            raise_error("url should not include the scheme")
        scheme = "https://" if ssl else "http://"
        self._url = scheme + url
        self._verbose = verbose
        connector = aiohttp.TCPConnector(limit=conn_limit, ssl=ssl_context if ssl else False)
        self._session = aiohttp.ClientSession(
            connector=connector, timeout=aiohttp.ClientTimeout(total=conn_timeout),
            auto_decompress=False,
            # aiohttp otherwise advertises gzip itself; compression is
            # opt-in via response_compression_algorithm (we decompress
            # manually in InferResult)
            headers={"Accept-Encoding": "identity"},
        )

    async def __aenter__(self):
        return self

    async def __aexit__(self, type, value, traceback):
        await self.close()

    async def close(self):
        await self._session.close()

    def _validate_headers(self, headers):
        if headers is not None:
            for key in headers.keys():
                if key.lower() == "transfer-encoding":
                    raise_error(
                        "Transfer-Encoding is not allowed as a request header"
                    )

    async def _get(self, request_uri, headers=None, query_params=None):
        self._validate_headers(headers)
        req = Request(dict(headers) if headers else {})
        self._call_plugin(req)
        url = self._url + "/" + request_uri
        if self._verbose:
            print(f"GET {url}")
        async with self._session.get(
            url, headers=req.headers, params=query_params
        ) as resp:
            body = await resp.read()
            return _Response(resp.status, resp.headers, body)

    async def _post(self, request_uri, request_body=None, headers=None,
                    query_params=None):
        self._validate_headers(headers)
        req = Request(dict(headers) if headers else {})
        self._call_plugin(req)
        url = self._url + "/" + request_uri
        if self._verbose:
            print(f"POST {url}")
        async with self._session.post(
            url, data=request_body, headers=req.headers, params=query_params
        ) as resp:
            body = await resp.read()
            return _Response(resp.status, resp.headers, body)

    # ---- health / metadata ----

    async def is_server_live(self, headers=None, query_params=None):
        response = await self._get("v2/health/live", headers, query_params)
        return response.status == 200

    async def is_server_ready(self, headers=None, query_params=None):
        response = await self._get("v2/health/ready", headers, query_params)
        return response.status == 200

    async def is_model_ready(self, model_name, model_version="", headers=None,
                             query_params=None):
        if model_version != "":
            uri = "v2/models/{}/versions/{}/ready".format(
                urllib.parse.quote(model_name), model_version
            )
        else:
            uri = "v2/models/{}/ready".format(urllib.parse.quote(model_name))
        response = await self._get(uri, headers, query_params)
        return response.status == 200

    async def get_server_metadata(self, headers=None, query_params=None):
        response = await self._get("v2", headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    async def get_model_metadata(self, model_name, model_version="", headers=None,
                                 query_params=None):
        if model_version != "":
            uri = "v2/models/{}/versions/{}".format(
                urllib.parse.quote(model_name), model_version
            )
        else:
            uri = "v2/models/{}".format(urllib.parse.quote(model_name))
        response = await self._get(uri, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    async def get_model_config(self, model_name, model_version="", headers=None,
                               query_params=None):
        if model_version != "":
            uri = "v2/models/{}/versions/{}/config".format(
                urllib.parse.quote(model_name), model_version
            )
        else:
            uri = "v2/models/{}/config".format(urllib.parse.quote(model_name))
        response = await self._get(uri, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    # ---- repository ----

    async def get_model_repository_index(self, headers=None, query_params=None):
        response = await self._post("v2/repository/index", None, headers,
                                    query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    async def load_model(self, model_name, headers=None, query_params=None,
                         config=None, files=None):
        uri = "v2/repository/models/{}/load".format(urllib.parse.quote(model_name))
        load_request = {}
        if config is not None:
            load_request.setdefault("parameters", {})["config"] = config
        if files is not None:
            for path, content in files.items():
                load_request.setdefault("parameters", {})[path] = base64.b64encode(
                    content
                ).decode("ascii")
        response = await self._post(uri, json.dumps(load_request), headers,
                                    query_params)
        _raise_if_error(response.status, response.read())

    async def unload_model(self, model_name, headers=None, query_params=None,
                           unload_dependents=False):
        uri = "v2/repository/models/{}/unload".format(urllib.parse.quote(model_name))
        body = json.dumps({"parameters": {"unload_dependents": unload_dependents}})
        response = await self._post(uri, body, headers, query_params)
        _raise_if_error(response.status, response.read())

    # ---- statistics / settings ----

    async def get_inference_statistics(self, model_name="", model_version="",
                                       headers=None, query_params=None):
        if model_name != "":
            if model_version != "":
                uri = "v2/models/{}/versions/{}/stats".format(
                    urllib.parse.quote(model_name), model_version
                )
            else:
                uri = "v2/models/{}/stats".format(urllib.parse.quote(model_name))
        else:
            uri = "v2/models/stats"
        response = await self._get(uri, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    async def update_trace_settings(self, model_name=None, settings={},
                                    headers=None, query_params=None):
        if model_name is not None and model_name != "":
            uri = "v2/models/{}/trace/setting".format(urllib.parse.quote(model_name))
        else:
            uri = "v2/trace/setting"
        response = await self._post(uri, json.dumps(settings), headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    async def get_trace_settings(self, model_name=None, headers=None,
                                 query_params=None):
        if model_name is not None and model_name != "":
            uri = "v2/models/{}/trace/setting".format(urllib.parse.quote(model_name))
        else:
            uri = "v2/trace/setting"
        response = await self._get(uri, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    async def update_log_settings(self, settings, headers=None, query_params=None):
        response = await self._post("v2/logging", json.dumps(settings), headers,
                                    query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    async def get_log_settings(self, headers=None, query_params=None):
        response = await self._get("v2/logging", headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    # ---- shared memory ----

    async def get_system_shared_memory_status(self, region_name="", headers=None,
                                              query_params=None):
        if region_name != "":
            uri = "v2/systemsharedmemory/region/{}/status".format(
                urllib.parse.quote(region_name)
            )
        else:
            uri = "v2/systemsharedmemory/status"
        response = await self._get(uri, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    async def register_system_shared_memory(self, name, key, byte_size, offset=0,
                                            headers=None, query_params=None):
        uri = "v2/systemsharedmemory/region/{}/register".format(
            urllib.parse.quote(name)
        )
        body = json.dumps({"key": key, "offset": offset, "byte_size": byte_size})
        response = await self._post(uri, body, headers, query_params)
        _raise_if_error(response.status, response.read())

    async def unregister_system_shared_memory(self, name="", headers=None,
                                              query_params=None):
        if name != "":
            uri = "v2/systemsharedmemory/region/{}/unregister".format(
                urllib.parse.quote(name)
            )
        else:
            uri = "v2/systemsharedmemory/unregister"
        response = await self._post(uri, None, headers, query_params)
        _raise_if_error(response.status, response.read())

    async def get_cuda_shared_memory_status(self, region_name="", headers=None,
                                            query_params=None):
        if region_name != "":
            uri = "v2/cudasharedmemory/region/{}/status".format(
                urllib.parse.quote(region_name)
            )
        else:
            uri = "v2/cudasharedmemory/status"
        response = await self._get(uri, headers, query_params)
        _raise_if_error(response.status, response.read())
        return json.loads(response.read())

    async def register_cuda_shared_memory(self, name, raw_handle, device_id,
                                          byte_size, headers=None,
                                          query_params=None):
        uri = "v2/cudasharedmemory/region/{}/register".format(
            urllib.parse.quote(name)
        )
        if isinstance(raw_handle, bytes):
            b64_handle = raw_handle.decode("ascii")
        else:
            b64_handle = raw_handle
        body = json.dumps({
            "raw_handle": {"b64": b64_handle},
            "device_id": device_id,
            "byte_size": byte_size,
        })
        response = await self._post(uri, body, headers, query_params)
        _raise_if_error(response.status, response.read())

    async def unregister_cuda_shared_memory(self, name="", headers=None,
                                            query_params=None):
        if name != "":
            uri = "v2/cudasharedmemory/region/{}/unregister".format(
                urllib.parse.quote(name)
            )
        else:
            uri = "v2/cudasharedmemory/unregister"
        response = await self._post(uri, None, headers, query_params)
        _raise_if_error(response.status, response.read())

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
        headers=None,
        query_params=None,
        request_compression_algorithm=None,
        response_compression_algorithm=None,
        parameters=None,
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
        hdrs = dict(headers) if headers else {}
        if request_compression_algorithm == "gzip":
            hdrs["Content-Encoding"] = "gzip"
            request_body = gzip.compress(request_body)
        elif request_compression_algorithm == "deflate":
            hdrs["Content-Encoding"] = "deflate"
            request_body = zlib.compress(request_body)
        if response_compression_algorithm in ("gzip", "deflate"):
            hdrs["Accept-Encoding"] = response_compression_algorithm
        if json_size is not None:
            hdrs["Inference-Header-Content-Length"] = str(json_size)
        if model_version != "":
            uri = "v2/models/{}/versions/{}/infer".format(
                urllib.parse.quote(model_name), model_version
            )
        else:
            uri = "v2/models/{}/infer".format(urllib.parse.quote(model_name))
        response = await self._post(uri, request_body, hdrs, query_params)
        _raise_if_error(response.status, response.read())
        return InferResult(response, self._verbose)
