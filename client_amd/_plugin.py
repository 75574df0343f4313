"""Client plugin ABC (reference: tritonclient/_plugin.py:31-48)."""

import abc


class InferenceServerClientPlugin(abc.ABC):
    """A plugin is called before every network operation with the Request
    about to be sent; it mutates ``request.headers`` in place (e.g. to
    inject auth headers)."""

    @abc.abstractmethod
    def __call__(self, request):
        pass
