"""model_config_pb2 compatibility module.

The reference wheel ships a protoc-generated ``tritonclient.grpc.
model_config_pb2``; here the same message classes come from the
runtime-built schema (``_proto.py``). Enum values are exposed as
module-level constants the way generated code does (``TYPE_FP32``,
``KIND_GPU``, ...).
"""

from ._proto import _ENUMS, _MESSAGES, _pool, service_pb2

_MODEL_PREFIXES = ("Model", "Batch")

# top-level model-config message classes (nested ones hang off their
# parents, as in generated code)
for _name in _MESSAGES:
    if "." not in _name and _name.startswith(_MODEL_PREFIXES):
        globals()[_name] = getattr(service_pb2, _name)

# module-level enum value constants (proto3 scoping puts top-level enum
# values in the file scope; generated pb2 mirrors that)
for _ename, _values in _ENUMS.items():
    if "." not in _ename:
        globals()[_ename] = getattr(service_pb2, _ename, None)
        for _i, _v in enumerate(_values):
            globals()[_v] = _i

del _name, _ename, _values, _i, _v
