"""Render proto/grpc_service.proto from the runtime schema.

`_proto.py`'s _MESSAGES/_ENUMS/RPCS tables are the source of truth for
the wire schema (no protoc in this stack); this module renders them as
canonical .proto text so third-party languages (go/java/js) can protoc
their own stubs. `python -m client_amd.grpc._proto_gen` rewrites the
vendored file; tests assert it is in sync.
"""

from . import _proto

_HEADER = """\
// KServe Predict Protocol v2 — gRPC schema served and consumed by
// client_amd (generated from client_amd/grpc/_proto.py, the runtime
// source of truth; wire-compatible with tritonclient/KServe-v2).
// Use with protoc for third-party language stubs (go/java/js).
// Regenerate: python -m client_amd.grpc._proto_gen
syntax = "proto3";

package inference;

"""


def _children(prefix):
    """Direct child message names of a dotted prefix ('' = top level)."""
    out = []
    for name in _proto._MESSAGES:
        parent, _, _ = name.rpartition(".")
        if parent == prefix:
            out.append(name)
    return out


def _child_enums(prefix):
    return [n for n in _proto._ENUMS if n.rpartition(".")[0] == prefix]


def _render_field(spec, indent):
    name, number, ftype = spec[0], spec[1], spec[2]
    flags = spec[3] if len(spec) > 3 else ""
    pad = " " * indent
    if ftype.startswith("map:"):
        _, keytype, valspec = ftype.split(":", 2)
        val = valspec[4:] if valspec.startswith("msg:") else valspec
        return f"{pad}map<{keytype}, {val}> {name} = {number};"
    if ftype.startswith("msg:"):
        tname = ftype[4:]
    elif ftype.startswith("enum:"):
        tname = ftype[5:]
    else:
        tname = ftype
    rep = "repeated " if ("r" in flags and not flags.startswith("o")) else ""
    return f"{pad}{rep}{tname} {name} = {number};"


def _render_enum(full_name, indent):
    pad = " " * indent
    short = full_name.rpartition(".")[2]
    lines = [f"{pad}enum {short} {{"]
    for i, v in enumerate(_proto._ENUMS[full_name]):
        lines.append(f"{pad}  {v} = {i};")
    lines.append(f"{pad}}}")
    return lines


def _render_message(full_name, indent):
    pad = " " * indent
    short = full_name.rpartition(".")[2]
    lines = [f"{pad}message {short} {{"]
    for e in _child_enums(full_name):
        lines.extend(_render_enum(e, indent + 2))
    for child in _children(full_name):
        lines.extend(_render_message(child, indent + 2))
    fields = _proto._MESSAGES[full_name]
    open_oneof = None
    for spec in fields:
        flags = spec[3] if len(spec) > 3 else ""
        group = flags[1:] if flags.startswith("o") else None
        if group != open_oneof:
            if open_oneof is not None:
                lines.append(f"{pad}  }}")
            if group is not None:
                lines.append(f"{pad}  oneof {group} {{")
            open_oneof = group
        lines.append(_render_field(spec, indent + (4 if group else 2)))
    if open_oneof is not None:
        lines.append(f"{pad}  }}")
    lines.append(f"{pad}}}")
    return lines


def render():
    out = [_HEADER + "service GRPCInferenceService {"]
    for rpc, (req, resp, streaming) in _proto.RPCS.items():
        req_n, resp_n = req.DESCRIPTOR.name, resp.DESCRIPTOR.name
        if streaming:
            out.append(
                f"  rpc {rpc}(stream {req_n}) returns (stream {resp_n}) {{}}"
            )
        else:
            out.append(f"  rpc {rpc}({req_n}) returns ({resp_n}) {{}}")
    out.append("}")
    out.append("")
    # Top-level enums first (DataType), then messages in table order.
    for e in _child_enums(""):
        out.extend(_render_enum(e, 0))
        out.append("")
    for m in _children(""):
        out.extend(_render_message(m, 0))
        out.append("")
    return "\n".join(out)


def main():
    import pathlib

    target = pathlib.Path(__file__).resolve().parents[2] / "proto" / (
        "grpc_service.proto"
    )
    target.write_text(render())
    print(f"wrote {target}")


if __name__ == "__main__":
    main()
