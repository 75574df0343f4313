"""Endpoint selection + retry: the capability surface of the reference
Java client's endpoint package and per-request retry count
(reference src/java/.../endpoint/AbstractEndpoint.java,
FixedEndpoint.java; retry loop InferenceServerClient.java:245-374) —
nothing equivalent existed in the reference C++/Python stacks, so this
is the one place the feature lives.

``MultiEndpointClient`` wraps one concrete client per URL and rotates
per request; on a transport error it retries against the next
endpoint(s) up to ``retries`` additional attempts.
"""

import itertools
import threading

from .utils import InferenceServerException

__all__ = [
    "FixedEndpoint",
    "RoundRobinEndpoint",
    "MultiEndpointClient",
]


class FixedEndpoint:
    """Always the same URL (Java FixedEndpoint)."""

    def __init__(self, url):
        self._url = url

    def get_next(self):
        return self._url

    def size(self):
        return 1

    def urls(self):
        return [self._url]


class RoundRobinEndpoint:
    """Rotate through a list of URLs, one per request (the Java
    AbstractEndpoint contract: getNext() called per request)."""

    def __init__(self, urls):
        if not urls:
            raise ValueError("at least one url required")
        self._urls = list(urls)
        self._it = itertools.cycle(self._urls)
        self._lock = threading.Lock()

    def get_next(self):
        with self._lock:
            return next(self._it)

    def size(self):
        return len(self._urls)

    def urls(self):
        return list(self._urls)


class MultiEndpointClient:
    """Protocol-agnostic rotating/failover client.

    Owns one concrete ``InferenceServerClient`` per URL (HTTP or gRPC,
    chosen by ``protocol``) and dispatches each call to the endpoint the
    policy picks; transport failures retry on subsequent endpoints up to
    ``retries`` extra attempts (Java retryCnt semantics).
    """

    def __init__(self, endpoint, protocol="http", retries=0,
                 client_factory=None, **client_kwargs):
        if isinstance(endpoint, (list, tuple)):
            endpoint = RoundRobinEndpoint(endpoint)
        elif isinstance(endpoint, str):
            endpoint = FixedEndpoint(endpoint)
        self._endpoint = endpoint
        self.retries = retries
        self.protocol = protocol  # lets InferRequestBuilder resolve IO types
        if client_factory is None:
            if protocol == "http":
                from .http import InferenceServerClient as factory
            elif protocol == "grpc":
                from .grpc import InferenceServerClient as factory
            else:
                raise ValueError(f"unknown protocol {protocol!r}")
            client_factory = factory
        self._clients = {
            url: client_factory(url, **client_kwargs)
            for url in endpoint.urls()
        }

    def client_for(self, url):
        return self._clients[url]

    def _call(self, method, *args, **kwargs):
        last_exc = None
        for _ in range(self.retries + 1):
            url = self._endpoint.get_next()
            client = self._clients[url]
            try:
                return getattr(client, method)(*args, **kwargs)
            except InferenceServerException as e:
                # an inference-level error from a healthy server is NOT
                # retried (it would fail everywhere); only transport
                # failures rotate (status None = network/connection)
                if e.status() is not None:
                    raise
                last_exc = e
            except (ConnectionError, OSError, TimeoutError) as e:
                last_exc = e
        if isinstance(last_exc, InferenceServerException):
            raise last_exc
        raise InferenceServerException(
            f"all {self.retries + 1} attempt(s) failed: {last_exc}"
        )

    # the client API surface, dispatched through the policy
    def infer(self, *args, **kwargs):
        return self._call("infer", *args, **kwargs)

    def async_infer(self, *args, **kwargs):
        return self._call("async_infer", *args, **kwargs)

    def is_server_live(self, *args, **kwargs):
        return self._call("is_server_live", *args, **kwargs)

    def is_server_ready(self, *args, **kwargs):
        return self._call("is_server_ready", *args, **kwargs)

    def is_model_ready(self, *args, **kwargs):
        return self._call("is_model_ready", *args, **kwargs)

    def get_server_metadata(self, *args, **kwargs):
        return self._call("get_server_metadata", *args, **kwargs)

    def get_model_metadata(self, *args, **kwargs):
        return self._call("get_model_metadata", *args, **kwargs)

    def get_model_config(self, *args, **kwargs):
        return self._call("get_model_config", *args, **kwargs)

    def get_inference_statistics(self, *args, **kwargs):
        return self._call("get_inference_statistics", *args, **kwargs)

    def close(self):
        for client in self._clients.values():
            client.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
