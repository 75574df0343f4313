"""Bi-directional streaming plumbing
(reference: tritonclient/grpc/_infer_stream.py:39-191).

_InferStream: a queue.Queue request pipe feeding the gRPC
request-iterator, plus a handler thread draining responses into the
user callback. _RequestIterator: the blocking generator gRPC consumes.
"""

import queue
import threading

import grpc

from ..utils import InferenceServerException
from ._infer_result import InferResult
from ._utils import get_cancelled_error, get_error_grpc


class _InferStream:
    def __init__(self, callback, verbose):
        self._callback = callback
        self._verbose = verbose
        self._request_queue = queue.Queue()
        self._handler = None
        self._active = True
        self._response_iterator = None

    def __del__(self):
        self.close(cancel_requests=True)

    def close(self, cancel_requests=False):
        if cancel_requests and self._response_iterator is not None:
            self._response_iterator.cancel()
            self._active = False
        if self._handler is not None:
            if not cancel_requests:
                self._request_queue.put(None)
            if self._handler.is_alive():
                self._handler.join()
            if self._verbose:
                print("stream stopped...")
            self._handler = None

    def _init_handler(self, response_iterator):
        self._response_iterator = response_iterator
        if self._handler is not None:
            raise InferenceServerException("Attempted to initialize already initialized InferStream")
        self._handler = threading.Thread(target=self._process_response)
        self._handler.start()
        if self._verbose:
            print("stream started...")

    def _enqueue_request(self, request):
        if not self._active:
            raise InferenceServerException(
                "The stream is no longer in valid state, the error detail "
                "is reported through provided callback. A new stream should "
                "be started after stopping the current stream."
            )
        self._request_queue.put(request)

    def _get_request(self):
        return self._request_queue.get()

    def is_active(self):
        return self._active

    def _process_response(self):
        """Drain the response iterator; each response fires the callback
        with (result, error) exactly-one-set semantics."""
        try:
            for response in self._response_iterator:
                if self._verbose:
                    print(response)
                result = error = None
                if response.error_message != "":
                    error = InferenceServerException(msg=response.error_message)
                else:
                    result = InferResult(response.infer_response)
                self._callback(result=result, error=error)
        except grpc.RpcError as rpc_error:
            # On stream breakage mark inactive and report
            self._active = False
            if rpc_error.code() == grpc.StatusCode.CANCELLED:
                error = get_cancelled_error()
            else:
                error = get_error_grpc(rpc_error)
            self._callback(result=None, error=error)


class _RequestIterator:
    """Blocking iterator over the stream's request queue; a ``None``
    sentinel ends the stream (reference _infer_stream.py:169-191)."""

    def __init__(self, stream):
        self._stream = stream

    def __iter__(self):
        return self

    def __next__(self):
        request = self._stream._get_request()
        if request is None:
            raise StopIteration
        return request
