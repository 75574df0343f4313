from client_amd.http.aio import *  # noqa: F401,F403
