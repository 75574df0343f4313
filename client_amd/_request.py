"""Request holder passed to plugins (reference: tritonclient/_request.py)."""


class Request:
    """Carries the mutable header map for an outgoing request.

    Parameters
    ----------
    headers : dict
        HTTP headers / gRPC metadata for the request.
    """

    def __init__(self, headers):
        self.headers = headers
