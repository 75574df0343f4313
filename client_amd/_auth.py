"""Auth plugins (reference: tritonclient/_auth.py:33-45)."""

import base64

from ._plugin import InferenceServerClientPlugin


class BasicAuth(InferenceServerClientPlugin):
    """HTTP basic auth: injects ``authorization: Basic <b64(user:pass)>``."""

    def __init__(self, username, password):
        creds = f"{username}:{password}".encode("utf-8")
        self._auth_header = "Basic " + base64.b64encode(creds).decode("ascii")

    def __call__(self, request):
        request.headers["authorization"] = self._auth_header
