""".proto → runtime message classes, without protoc.

This environment has the ``google.protobuf`` runtime but neither protoc nor
grpcio-tools, so this module compiles the vendored contract files itself:
tokenize → parse → ``FileDescriptorProto`` → ``DescriptorPool.Add`` →
``message_factory.GetMessageClass``. Covers the proto3 subset the contract
uses: packages, imports of well-known types, enums, (nested) messages,
oneofs, proto3 ``optional`` (synthetic oneofs), ``map<>`` fields,
``reserved``, field/enum options (parsed, ignored), and services with
streaming RPCs.

Parity: replaces the reference's protoc + custom wrapper plugin
(/root/reference/py/compile_protos.py, py/protoc_plugin/plugin.py).
"""

from __future__ import annotations

import os
import re
from typing import Any, Optional

from google.protobuf import descriptor_pb2

SCALARS = {
    "double": descriptor_pb2.FieldDescriptorProto.TYPE_DOUBLE,
    "float": descriptor_pb2.FieldDescriptorProto.TYPE_FLOAT,
    "int64": descriptor_pb2.FieldDescriptorProto.TYPE_INT64,
    "uint64": descriptor_pb2.FieldDescriptorProto.TYPE_UINT64,
    "int32": descriptor_pb2.FieldDescriptorProto.TYPE_INT32,
    "fixed64": descriptor_pb2.FieldDescriptorProto.TYPE_FIXED64,
    "fixed32": descriptor_pb2.FieldDescriptorProto.TYPE_FIXED32,
    "bool": descriptor_pb2.FieldDescriptorProto.TYPE_BOOL,
    "string": descriptor_pb2.FieldDescriptorProto.TYPE_STRING,
    "bytes": descriptor_pb2.FieldDescriptorProto.TYPE_BYTES,
    "uint32": descriptor_pb2.FieldDescriptorProto.TYPE_UINT32,
    "sfixed32": descriptor_pb2.FieldDescriptorProto.TYPE_SFIXED32,
    "sfixed64": descriptor_pb2.FieldDescriptorProto.TYPE_SFIXED64,
    "sint32": descriptor_pb2.FieldDescriptorProto.TYPE_SINT32,
    "sint64": descriptor_pb2.FieldDescriptorProto.TYPE_SINT64,
}

# well-known types provided by the installed runtime
WELL_KNOWN_MESSAGES = {
    ".google.protobuf.Any", ".google.protobuf.Empty", ".google.protobuf.Struct",
    ".google.protobuf.Value", ".google.protobuf.ListValue",
    ".google.protobuf.Timestamp", ".google.protobuf.Duration",
    ".google.protobuf.DoubleValue", ".google.protobuf.FloatValue",
    ".google.protobuf.Int64Value", ".google.protobuf.UInt64Value",
    ".google.protobuf.Int32Value", ".google.protobuf.UInt32Value",
    ".google.protobuf.BoolValue", ".google.protobuf.StringValue",
    ".google.protobuf.BytesValue", ".google.protobuf.FieldMask",
}
WELL_KNOWN_ENUMS = {".google.protobuf.NullValue"}

_TOKEN_RE = re.compile(
    r"""
    \s+
  | //[^\n]*
  | /\*.*?\*/
  | "(?:[^"\\]|\\.)*"
  | '(?:[^'\\]|\\.)*'
  | [A-Za-z_][A-Za-z0-9_.]*
  | -?\d+(?:\.\d+)?
  | [{}()\[\];=,<>]
    """,
    re.VERBOSE | re.DOTALL,
)


def tokenize(text: str) -> list[str]:
    tokens = []
    pos = 0
    n = len(text)
    while pos < n:
        m = _TOKEN_RE.match(text, pos)
        if m is None:
            raise SyntaxError(f"proto tokenizer stuck at {text[pos:pos+40]!r}")
        tok = m.group(0)
        pos = m.end()
        if tok[0].isspace() or tok.startswith("//") or tok.startswith("/*"):
            continue
        tokens.append(tok)
    return tokens


class _Parser:
    def __init__(self, tokens: list[str]):
        self.toks = tokens
        self.i = 0

    def peek(self) -> Optional[str]:
        return self.toks[self.i] if self.i < len(self.toks) else None

    def next(self) -> str:
        tok = self.toks[self.i]
        self.i += 1
        return tok

    def expect(self, tok: str) -> None:
        got = self.next()
        if got != tok:
            raise SyntaxError(f"expected {tok!r}, got {got!r} (at #{self.i})")

    def skip_to_matching_brace(self) -> None:
        """Consume a balanced {...} block (used for option bodies)."""
        depth = 0
        while True:
            tok = self.next()
            if tok == "{":
                depth += 1
            elif tok == "}":
                depth -= 1
                if depth == 0:
                    return

    def skip_statement(self) -> None:
        """Consume tokens through the next ';' or balanced '{...}'."""
        while True:
            tok = self.next()
            if tok == ";":
                return
            if tok == "{":
                self.i -= 1
                self.skip_to_matching_brace()
                return

    def skip_field_options(self) -> None:
        """Consume a [...] options annotation."""
        depth = 0
        while True:
            tok = self.next()
            if tok == "[":
                depth += 1
            elif tok == "]":
                depth -= 1
                if depth == 0:
                    return


def parse_file(text: str, file_name: str) -> descriptor_pb2.FileDescriptorProto:
    p = _Parser(tokenize(text))
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = file_name
    fdp.syntax = "proto3"

    while p.peek() is not None:
        tok = p.next()
        if tok == "syntax":
            p.expect("=")
            p.next()  # "proto3"
            p.expect(";")
        elif tok == "package":
            fdp.package = p.next()
            p.expect(";")
        elif tok == "import":
            dep = p.next()
            if dep in ("public", "weak"):
                dep = p.next()
            fdp.dependency.append(dep.strip("\"'"))
            p.expect(";")
        elif tok == "option":
            p.skip_statement()
        elif tok == "message":
            fdp.message_type.append(_parse_message(p))
        elif tok == "enum":
            fdp.enum_type.append(_parse_enum(p))
        elif tok == "service":
            fdp.service.append(_parse_service(p))
        elif tok == ";":
            continue
        else:
            raise SyntaxError(f"unexpected top-level token {tok!r}")
    return fdp


def _parse_enum(p: _Parser) -> descriptor_pb2.EnumDescriptorProto:
    enum = descriptor_pb2.EnumDescriptorProto()
    enum.name = p.next()
    p.expect("{")
    while True:
        tok = p.next()
        if tok == "}":
            return enum
        if tok == "option":
            p.skip_statement()
            continue
        if tok == "reserved":
            p.skip_statement()
            continue
        value = enum.value.add()
        value.name = tok
        p.expect("=")
        value.number = int(p.next())
        if p.peek() == "[":
            p.skip_field_options()
        p.expect(";")


def _parse_message(p: _Parser) -> descriptor_pb2.DescriptorProto:
    msg = descriptor_pb2.DescriptorProto()
    msg.name = p.next()
    p.expect("{")
    _parse_message_body(p, msg)
    return msg


def _parse_message_body(p: _Parser, msg: descriptor_pb2.DescriptorProto) -> None:
    proto3_optional_fields: list[descriptor_pb2.FieldDescriptorProto] = []
    while True:
        tok = p.next()
        if tok == "}":
            break
        if tok == ";":
            continue
        if tok == "message":
            msg.nested_type.append(_parse_message(p))
        elif tok == "enum":
            msg.enum_type.append(_parse_enum(p))
        elif tok in ("option", "reserved", "extensions"):
            p.skip_statement()
        elif tok == "oneof":
            oneof = msg.oneof_decl.add()
            oneof.name = p.next()
            oneof_index = len(msg.oneof_decl) - 1
            p.expect("{")
            while True:
                t2 = p.next()
                if t2 == "}":
                    break
                if t2 == "option":
                    p.skip_statement()
                    continue
                field = _parse_field(p, t2, label_tok=None)
                field.oneof_index = oneof_index
                msg.field.append(field)
        elif tok == "map":
            # map<K, V> name = N  →  repeated synthetic MapEntry message
            p.expect("<")
            key_type = p.next()
            p.expect(",")
            val_type = p.next()
            p.expect(">")
            name = p.next()
            p.expect("=")
            number = int(p.next())
            if p.peek() == "[":
                p.skip_field_options()
            p.expect(";")
            entry_name = "".join(w.capitalize() for w in name.split("_")) + "Entry"
            entry = msg.nested_type.add()
            entry.name = entry_name
            entry.options.map_entry = True
            kf = entry.field.add()
            kf.name, kf.number = "key", 1
            kf.label = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
            kf.type = SCALARS[key_type]
            vf = entry.field.add()
            vf.name, vf.number = "value", 2
            vf.label = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
            if val_type in SCALARS:
                vf.type = SCALARS[val_type]
            else:
                vf.type = descriptor_pb2.FieldDescriptorProto.TYPE_MESSAGE
                vf.type_name = val_type  # resolved in the resolve pass
            field = msg.field.add()
            field.name = name
            field.number = number
            field.label = descriptor_pb2.FieldDescriptorProto.LABEL_REPEATED
            field.type = descriptor_pb2.FieldDescriptorProto.TYPE_MESSAGE
            field.type_name = entry_name  # sibling scope; resolve pass fixes
        elif tok in ("optional", "repeated", "required"):
            field = _parse_field(p, p.next(), label_tok=tok)
            msg.field.append(field)
            if tok == "optional":
                # append COPIES the message: remember the index, mutate later
                proto3_optional_fields.append(len(msg.field) - 1)
        else:
            msg.field.append(_parse_field(p, tok, label_tok=None))
    # proto3 optional → one synthetic oneof per field, after the real ones
    for field_idx in proto3_optional_fields:
        field = msg.field[field_idx]
        oneof = msg.oneof_decl.add()
        oneof.name = "_" + field.name
        field.oneof_index = len(msg.oneof_decl) - 1
        field.proto3_optional = True


def _parse_field(
    p: _Parser, type_tok: str, label_tok: Optional[str]
) -> descriptor_pb2.FieldDescriptorProto:
    field = descriptor_pb2.FieldDescriptorProto()
    if label_tok == "repeated":
        field.label = descriptor_pb2.FieldDescriptorProto.LABEL_REPEATED
    else:
        field.label = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
    if type_tok in SCALARS:
        field.type = SCALARS[type_tok]
    else:
        field.type_name = type_tok  # message or enum: resolve pass decides
    field.name = p.next()
    p.expect("=")
    field.number = int(p.next())
    if p.peek() == "[":
        p.skip_field_options()
    p.expect(";")
    return field


def _parse_service(p: _Parser) -> descriptor_pb2.ServiceDescriptorProto:
    svc = descriptor_pb2.ServiceDescriptorProto()
    svc.name = p.next()
    p.expect("{")
    while True:
        tok = p.next()
        if tok == "}":
            return svc
        if tok == "option":
            p.skip_statement()
            continue
        assert tok == "rpc", f"unexpected {tok!r} in service"
        m = svc.method.add()
        m.name = p.next()
        p.expect("(")
        t = p.next()
        if t == "stream":
            m.client_streaming = True
            t = p.next()
        m.input_type = t
        p.expect(")")
        p.expect("returns")
        p.expect("(")
        t = p.next()
        if t == "stream":
            m.server_streaming = True
            t = p.next()
        m.output_type = t
        p.expect(")")
        if p.peek() == "{":
            p.skip_to_matching_brace()
        elif p.peek() == ";":
            p.next()


# ---------------------------------------------------------------------------
# name resolution
# ---------------------------------------------------------------------------


def _collect_symbols(fdp: descriptor_pb2.FileDescriptorProto, table: dict) -> None:
    pkg = "." + fdp.package if fdp.package else ""

    def walk(msg: descriptor_pb2.DescriptorProto, scope: str) -> None:
        fqn = scope + "." + msg.name
        table[fqn] = "message"
        for nested in msg.nested_type:
            walk(nested, fqn)
        for enum in msg.enum_type:
            table[fqn + "." + enum.name] = "enum"

    for msg in fdp.message_type:
        walk(msg, pkg)
    for enum in fdp.enum_type:
        table[pkg + "." + enum.name] = "enum"


def _resolve_name(ref: str, scope: str, table: dict) -> str:
    """C++-style resolution: innermost scope outward, then absolute."""
    if ref.startswith("."):
        if ref in table:
            return ref
        raise KeyError(f"unresolved absolute type {ref}")
    parts = scope.split(".") if scope else [""]
    while parts:
        candidate = ".".join(parts) + "." + ref
        if not candidate.startswith("."):
            candidate = "." + candidate
        if candidate in table:
            return candidate
        parts.pop()
    candidate = "." + ref
    if candidate in table:
        return candidate
    raise KeyError(f"unresolved type {ref!r} in scope {scope!r}")


def _resolve_file(fdp: descriptor_pb2.FileDescriptorProto, table: dict) -> None:
    pkg = "." + fdp.package if fdp.package else ""

    def fix_field(field: descriptor_pb2.FieldDescriptorProto, scope: str) -> None:
        if not field.type_name:
            return
        fqn = _resolve_name(field.type_name, scope, table)
        field.type_name = fqn
        if not field.type:
            field.type = (
                descriptor_pb2.FieldDescriptorProto.TYPE_ENUM
                if table[fqn] == "enum"
                else descriptor_pb2.FieldDescriptorProto.TYPE_MESSAGE
            )

    def walk(msg: descriptor_pb2.DescriptorProto, scope: str) -> None:
        fqn = scope + "." + msg.name
        for field in msg.field:
            fix_field(field, fqn)
        for nested in msg.nested_type:
            walk(nested, fqn)

    for msg in fdp.message_type:
        walk(msg, pkg)
    for svc in fdp.service:
        for m in svc.method:
            m.input_type = _resolve_name(m.input_type, pkg, table)
            m.output_type = _resolve_name(m.output_type, pkg, table)


# ---------------------------------------------------------------------------
# loading
# ---------------------------------------------------------------------------

_CACHE: dict[str, Any] = {}


class ProtoModule:
    """Namespace of generated message classes + enums for one .proto file
    (the moral equivalent of a *_pb2 module)."""

    def __init__(self, fdp: descriptor_pb2.FileDescriptorProto, pool: Any):
        from google.protobuf import message_factory

        self._fdp = fdp
        self._pool = pool
        self.DESCRIPTOR = pool.FindFileByName(fdp.name)
        pkg = fdp.package
        for msg in fdp.message_type:
            full = f"{pkg}.{msg.name}" if pkg else msg.name
            setattr(self, msg.name, message_factory.GetMessageClass(
                pool.FindMessageTypeByName(full)
            ))
        for enum in fdp.enum_type:
            full = f"{pkg}.{enum.name}" if pkg else enum.name
            enum_desc = pool.FindEnumTypeByName(full)
            wrapper = _EnumNamespace(enum_desc)
            setattr(self, enum.name, wrapper)
            for value in enum_desc.values:  # proto3 enums leak values to file scope
                setattr(self, value.name, value.number)
        self.SERVICES = {svc.name: svc for svc in fdp.service}


class _EnumNamespace:
    def __init__(self, enum_desc: Any):
        self.DESCRIPTOR = enum_desc
        for value in enum_desc.values:
            setattr(self, value.name, value.number)

    def Name(self, number: int) -> str:
        return self.DESCRIPTOR.values_by_number[number].name

    def Value(self, name: str) -> int:
        return self.DESCRIPTOR.values_by_name[name].number


def compile_proto_files(paths: list[tuple[str, str]]) -> dict[str, ProtoModule]:
    """paths: [(pool_file_name, fs_path)] in dependency order."""
    # well-known descriptors register themselves on import
    from google.protobuf import (  # noqa: F401
        any_pb2, duration_pb2, empty_pb2, struct_pb2, timestamp_pb2, wrappers_pb2,
    )
    from google.protobuf import descriptor_pool

    pool = descriptor_pool.Default()
    table: dict[str, str] = {}
    for name in WELL_KNOWN_MESSAGES:
        table[name] = "message"
    for name in WELL_KNOWN_ENUMS:
        table[name] = "enum"

    fdps = []
    for pool_name, fs_path in paths:
        with open(fs_path) as f:
            fdp = parse_file(f.read(), pool_name)
        _collect_symbols(fdp, table)
        fdps.append(fdp)
    out = {}
    for fdp in fdps:
        _resolve_file(fdp, table)
        try:
            pool.Add(fdp)
        except Exception as exc:  # already registered in this process
            if "duplicate" not in str(exc).lower():
                raise
        out[fdp.name] = ProtoModule(fdp, pool)
    return out


def load() -> tuple[ProtoModule, ProtoModule]:
    """(api_pb2-equivalent, task_command_router_pb2-equivalent)."""
    if "api" not in _CACHE:
        here = os.path.dirname(os.path.abspath(__file__))
        modules = compile_proto_files(
            [
                ("modal_proto/api.proto", os.path.join(here, "api.proto")),
                (
                    "modal_proto/task_command_router.proto",
                    os.path.join(here, "task_command_router.proto"),
                ),
            ]
        )
        _CACHE["api"] = modules["modal_proto/api.proto"]
        _CACHE["router"] = modules["modal_proto/task_command_router.proto"]
    return _CACHE["api"], _CACHE["router"]
