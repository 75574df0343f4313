"""Base class shared by all protocol clients.

Single-slot plugin registry with a hook invoked before every network op
(reference: tritonclient/_client.py:31-85).
"""

from ._plugin import InferenceServerClientPlugin


class InferenceServerClientBase:
    def __init__(self):
        self._plugin = None

    def _call_plugin(self, request):
        """Invoke the registered plugin (if any) on the outgoing request."""
        if self._plugin is not None:
            self._plugin(request)

    def register_plugin(self, plugin):
        """Register a plugin; only one may be active at a time."""
        if not isinstance(plugin, InferenceServerClientPlugin):
            raise ValueError("plugin must be an InferenceServerClientPlugin")
        if self._plugin is None:
            self._plugin = plugin
        else:
            raise ValueError("A plugin is already registered. Unregister first.")

    def unregister_plugin(self):
        """Unregister the active plugin."""
        if self._plugin is None:
            raise ValueError("No plugin is registered.")
        self._plugin = None

    def plugin(self):
        """Return the currently registered plugin (or None)."""
        return self._plugin
