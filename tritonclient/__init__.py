"""tritonclient — drop-in alias for client_amd.

User code written against NVIDIA's ``tritonclient`` SDK runs unmodified
on the MI355X-native stack: every subpackage re-exports the client_amd
implementation (reference package layout:
/root/reference/src/python/library/tritonclient/__init__.py).
"""
