"""protoc-lite: parse .proto files into FileDescriptorProto objects.

Purpose ("what's missing" #1): every byte-compatibility
proof in round 1 ultimately rode on descriptors hand-declared in
wire/schema.py — one shared misreading of the reference's .proto files
would pass every test. This tool closes the loop by *mechanically*
deriving descriptors from the reference's own .proto sources
(/root/reference/protobuf_srcs/**.proto, read as protocol-definition
data), registering them in a SEPARATE descriptor pool, and letting tests
differential-check our schema and our native C++ codec against message
classes built from them.

The container has no protoc/grpcio-tools (reference setup.py:41-49 runs
protoc at build time), so this is a from-scratch recursive-descent parser
for the proto3 subset those files use: package/import/option, nested
message/enum, repeated/optional fields, map<>, oneof, reserved, services
(recorded but unused). Output can be serialized as a FileDescriptorSet
so the oracle also works where the reference tree is absent
(tests/fixtures/reference_descriptor_set.binpb).
"""
from __future__ import annotations

import re
from typing import Dict, List, Optional, Tuple

from google.protobuf import descriptor_pb2

_F = descriptor_pb2.FieldDescriptorProto

SCALAR_TYPES = {
    "double": _F.TYPE_DOUBLE, "float": _F.TYPE_FLOAT,
    "int64": _F.TYPE_INT64, "uint64": _F.TYPE_UINT64,
    "int32": _F.TYPE_INT32, "fixed64": _F.TYPE_FIXED64,
    "fixed32": _F.TYPE_FIXED32, "bool": _F.TYPE_BOOL,
    "string": _F.TYPE_STRING, "bytes": _F.TYPE_BYTES,
    "uint32": _F.TYPE_UINT32, "sfixed32": _F.TYPE_SFIXED32,
    "sfixed64": _F.TYPE_SFIXED64, "sint32": _F.TYPE_SINT32,
    "sint64": _F.TYPE_SINT64,
}

# well-known types assumed present in the target pool
WELL_KNOWN = {
    "google.protobuf.Any", "google.protobuf.Int64Value",
    "google.protobuf.Int32Value", "google.protobuf.UInt64Value",
    "google.protobuf.UInt32Value", "google.protobuf.DoubleValue",
    "google.protobuf.FloatValue", "google.protobuf.BoolValue",
    "google.protobuf.StringValue", "google.protobuf.BytesValue",
    "google.protobuf.Duration", "google.protobuf.Timestamp",
    "google.protobuf.Struct", "google.protobuf.Value",
    "google.protobuf.ListValue", "google.protobuf.FieldMask",
    "google.protobuf.Empty",
}


class Tokenizer:
    _TOKEN_RE = re.compile(
        r"""
        \s+
        | //[^\n]*
        | /\*.*?\*/
        | (?P<str>"(?:\\.|[^"\\])*")
        | (?P<sym>[{}()\[\];=,<>])
        | (?P<word>[A-Za-z0-9_.+-]+)
        """, re.VERBOSE | re.DOTALL)

    def __init__(self, text: str):
        self.tokens: List[str] = []
        pos = 0
        while pos < len(text):
            m = self._TOKEN_RE.match(text, pos)
            if not m:
                raise SyntaxError(f"bad token at {text[pos:pos+40]!r}")
            pos = m.end()
            if m.lastgroup in ("str", "sym", "word"):
                self.tokens.append(m.group(m.lastgroup))
        self.i = 0

    def peek(self) -> Optional[str]:
        return self.tokens[self.i] if self.i < len(self.tokens) else None

    def next(self) -> str:
        tok = self.peek()
        if tok is None:
            raise SyntaxError("unexpected EOF")
        self.i += 1
        return tok

    def expect(self, tok: str) -> None:
        got = self.next()
        if got != tok:
            raise SyntaxError(f"expected {tok!r}, got {got!r}")

    def skip_balanced_braces(self) -> None:
        """consumes a '{ ... }' block"""
        self.expect("{")
        depth = 1
        while depth:
            t = self.next()
            if t == "{":
                depth += 1
            elif t == "}":
                depth -= 1

    def skip_statement(self) -> None:
        """consumes tokens until ';' or a balanced '{...}' block"""
        while True:
            t = self.next()
            if t == ";":
                return
            if t == "{":
                depth = 1
                while depth:
                    t2 = self.next()
                    if t2 == "{":
                        depth += 1
                    elif t2 == "}":
                        depth -= 1
                return


def _camel(field_name: str) -> str:
    return "".join(p.capitalize() for p in field_name.split("_"))


class ProtoFileParser:
    """One .proto file -> FileDescriptorProto (type names unresolved:
    message/enum-typed fields carry the written name in type_name with a
    leading '!'; resolve_types() fixes them up once all files are known).
    """

    def __init__(self, text: str, name: str):
        self.tz = Tokenizer(text)
        self.fdp = descriptor_pb2.FileDescriptorProto()
        self.fdp.name = name
        self.fdp.syntax = "proto3"

    def parse(self) -> descriptor_pb2.FileDescriptorProto:
        tz = self.tz
        while tz.peek() is not None:
            t = tz.next()
            if t == "syntax":
                tz.expect("=")
                s = tz.next()
                if s != '"proto3"':
                    raise SyntaxError(f"only proto3 supported, got {s}")
                tz.expect(";")
            elif t == "package":
                self.fdp.package = tz.next()
                tz.expect(";")
            elif t == "import":
                nxt = tz.next()
                if nxt in ("public", "weak"):
                    nxt = tz.next()
                self.fdp.dependency.append(nxt.strip('"'))
                tz.expect(";")
            elif t == "option":
                tz.skip_statement()
            elif t == "message":
                self.fdp.message_type.add().CopyFrom(self._parse_message())
            elif t == "enum":
                self.fdp.enum_type.add().CopyFrom(self._parse_enum())
            elif t == "service":
                self._parse_service()
            elif t == ";":
                pass
            else:
                raise SyntaxError(f"unexpected top-level token {t!r}")
        return self.fdp

    # -- message ---------------------------------------------------------
    def _parse_message(self) -> descriptor_pb2.DescriptorProto:
        tz = self.tz
        msg = descriptor_pb2.DescriptorProto()
        msg.name = tz.next()
        tz.expect("{")
        while True:
            t = tz.next()
            if t == "}":
                return msg
            if t == ";":
                continue
            if t == "message":
                msg.nested_type.add().CopyFrom(self._parse_message())
            elif t == "enum":
                msg.enum_type.add().CopyFrom(self._parse_enum())
            elif t == "oneof":
                oneof_name = tz.next()
                oneof_index = len(msg.oneof_decl)
                msg.oneof_decl.add().name = oneof_name
                tz.expect("{")
                while tz.peek() != "}":
                    if tz.peek() == ";":
                        tz.next()
                        continue
                    f = self._parse_field(tz.next(), msg)
                    f.oneof_index = oneof_index
                    msg.field.add().CopyFrom(f)
                tz.expect("}")
            elif t in ("reserved", "extensions", "option"):
                tz.skip_statement()
            elif t == "map":
                f = self._parse_map_field(msg)
                msg.field.add().CopyFrom(f)
            else:
                msg.field.add().CopyFrom(self._parse_field(t, msg))

    def _parse_field(self, first: str,
                     msg: descriptor_pb2.DescriptorProto
                     ) -> descriptor_pb2.FieldDescriptorProto:
        tz = self.tz
        f = descriptor_pb2.FieldDescriptorProto()
        f.label = _F.LABEL_OPTIONAL
        if first == "repeated":
            f.label = _F.LABEL_REPEATED
            first = tz.next()
        elif first == "optional":  # proto3 explicit presence
            f.proto3_optional = True
            first = tz.next()
        if first == "map":
            raise SyntaxError("map handled by caller")
        type_name = first
        if type_name in SCALAR_TYPES:
            f.type = SCALAR_TYPES[type_name]
        else:
            # message or enum; resolved later (marker prefix '!')
            f.type_name = "!" + type_name
        f.name = tz.next()
        tz.expect("=")
        f.number = int(tz.next())
        nxt = tz.next()
        if nxt == "[":  # field options, e.g. [packed = true], [lazy = true]
            while True:
                t2 = tz.next()
                if t2 == "]":
                    break
            tz.expect(";")
        elif nxt != ";":
            raise SyntaxError(f"bad field tail {nxt!r}")
        # proto3_optional needs a synthetic oneof
        if f.proto3_optional:
            f.oneof_index = len(msg.oneof_decl)
            msg.oneof_decl.add().name = "_" + f.name
        return f

    def _parse_map_field(self, msg: descriptor_pb2.DescriptorProto
                         ) -> descriptor_pb2.FieldDescriptorProto:
        tz = self.tz
        tz.expect("<")
        key_type = tz.next()
        tz.expect(",")
        value_type = tz.next()
        tz.expect(">")
        name = tz.next()
        tz.expect("=")
        number = int(tz.next())
        tz.expect(";")
        # synthesize the map entry message (descriptor.proto map encoding)
        entry = msg.nested_type.add()
        entry.name = _camel(name) + "Entry"
        entry.options.map_entry = True
        kf = entry.field.add()
        kf.name, kf.number, kf.label = "key", 1, _F.LABEL_OPTIONAL
        kf.type = SCALAR_TYPES[key_type]
        vf = entry.field.add()
        vf.name, vf.number, vf.label = "value", 2, _F.LABEL_OPTIONAL
        if value_type in SCALAR_TYPES:
            vf.type = SCALAR_TYPES[value_type]
        else:
            vf.type_name = "!" + value_type
        f = descriptor_pb2.FieldDescriptorProto()
        f.name = name
        f.number = number
        f.label = _F.LABEL_REPEATED
        f.type = _F.TYPE_MESSAGE
        f.type_name = "!" + entry.name  # resolves within this message scope
        return f

    # -- enum ------------------------------------------------------------
    def _parse_enum(self) -> descriptor_pb2.EnumDescriptorProto:
        tz = self.tz
        enum = descriptor_pb2.EnumDescriptorProto()
        enum.name = tz.next()
        tz.expect("{")
        while True:
            t = tz.next()
            if t == "}":
                return enum
            if t == ";":
                continue
            if t in ("option", "reserved"):
                tz.skip_statement()
                continue
            v = enum.value.add()
            v.name = t
            tz.expect("=")
            v.number = int(tz.next())
            nxt = tz.next()
            if nxt == "[":
                while tz.next() != "]":
                    pass
                tz.expect(";")
            elif nxt != ";":
                raise SyntaxError(f"bad enum value tail {nxt!r}")

    # -- service (recorded as names only; gRPC stubs are elsewhere) ------
    def _parse_service(self) -> None:
        tz = self.tz
        svc = self.fdp.service.add()
        svc.name = tz.next()
        tz.expect("{")
        while True:
            t = tz.next()
            if t == "}":
                return
            if t == "rpc":
                m = svc.method.add()
                m.name = tz.next()
                tz.expect("(")
                m.input_type = "!" + tz.next()
                tz.expect(")")
                tz.expect("returns")
                tz.expect("(")
                m.output_type = "!" + tz.next()
                tz.expect(")")
                nxt = tz.next()
                if nxt == "{":
                    depth = 1
                    while depth:
                        t2 = tz.next()
                        if t2 == "{":
                            depth += 1
                        elif t2 == "}":
                            depth -= 1
                elif nxt != ";":
                    raise SyntaxError(f"bad rpc tail {nxt!r}")
            elif t in ("option",):
                tz.skip_statement()


# ---------------------------------------------------------------------------
# cross-file type resolution
# ---------------------------------------------------------------------------

def _collect_names(fdp, scope: str, messages: set, enums: set) -> None:
    for m in fdp.message_type:
        _collect_msg(m, scope, messages, enums)
    for e in fdp.enum_type:
        enums.add(f"{scope}.{e.name}" if scope else e.name)


def _collect_msg(m, scope: str, messages: set, enums: set) -> None:
    full = f"{scope}.{m.name}" if scope else m.name
    messages.add(full)
    for nested in m.nested_type:
        _collect_msg(nested, full, messages, enums)
    for e in m.enum_type:
        enums.add(f"{full}.{e.name}")


def _resolve_name(written: str, scope: str, messages: set,
                  enums: set) -> Tuple[str, bool]:
    """C++-style scoping: try innermost enclosing scope outward.
    Returns (fully_qualified, is_message)."""
    if written.startswith("."):
        name = written[1:]
        return name, name in messages or name in WELL_KNOWN
    scopes = []
    parts = scope.split(".") if scope else []
    for i in range(len(parts), -1, -1):
        prefix = ".".join(parts[:i])
        scopes.append(f"{prefix}.{written}" if prefix else written)
    for cand in scopes:
        if cand in messages or cand in WELL_KNOWN:
            return cand, True
        if cand in enums:
            return cand, False
    raise NameError(f"cannot resolve type {written!r} in scope {scope!r}")


def resolve_types(fdps: List[descriptor_pb2.FileDescriptorProto]) -> None:
    messages = set(WELL_KNOWN)
    enums = set()
    for fdp in fdps:
        _collect_names(fdp, fdp.package, messages, enums)

    def fix_fields(m, scope):
        for f in m.field:
            if f.type_name.startswith("!"):
                written = f.type_name[1:]
                full, is_msg = _resolve_name(written, scope, messages,
                                             enums)
                f.type_name = "." + full
                if not f.HasField("type"):
                    f.type = _F.TYPE_MESSAGE if is_msg else _F.TYPE_ENUM
        for nested in m.nested_type:
            fix_fields(nested, f"{scope}.{nested.name}")

    for fdp in fdps:
        pkg = fdp.package
        for m in fdp.message_type:
            fix_fields(m, f"{pkg}.{m.name}" if pkg else m.name)
        for svc in fdp.service:
            for meth in svc.method:
                for attr in ("input_type", "output_type"):
                    written = getattr(meth, attr)
                    if written.startswith("!"):
                        full, _ = _resolve_name(written[1:], pkg, messages,
                                                enums)
                        setattr(meth, attr, "." + full)


# ---------------------------------------------------------------------------
# driver
# ---------------------------------------------------------------------------

def parse_files(root: str, rel_paths: List[str]
                ) -> List[descriptor_pb2.FileDescriptorProto]:
    """Parses `rel_paths` (in dependency order) under `root`, resolves
    types, returns FileDescriptorProtos ready for DescriptorPool.Add."""
    import os
    fdps = []
    for rel in rel_paths:
        with open(os.path.join(root, rel)) as fh:
            text = fh.read()
        fdps.append(ProtoFileParser(text, rel).parse())
    resolve_types(fdps)
    return fdps


# the exercised closure, in dependency order
REFERENCE_CLOSURE = [
    "tensorflow/core/framework/types.proto",
    "tensorflow/core/framework/tensor_shape.proto",
    "tensorflow/core/framework/resource_handle.proto",
    "tensorflow/core/framework/tensor.proto",
    "tensorflow/core/framework/versions.proto",
    "tensorflow/core/framework/attr_value.proto",
    "tensorflow/core/framework/node_def.proto",
    "tensorflow/core/framework/op_def.proto",
    "tensorflow/core/framework/function.proto",
    "tensorflow/core/framework/graph.proto",
    "tensorflow/core/framework/variable.proto",
    "tensorflow/core/protobuf/saver.proto",
    "tensorflow/core/protobuf/struct.proto",
    "tensorflow/core/protobuf/trackable_object_graph.proto",
    "tensorflow/core/protobuf/saved_object_graph.proto",
    "tensorflow/core/protobuf/meta_graph.proto",
    "tensorflow/core/example/feature.proto",
    "tensorflow/core/example/example.proto",
    # lib/core/error_codes.proto is an `import public` alias of this one
    "tensorflow/core/protobuf/error_codes.proto",
    "tensorflow_serving/apis/model.proto",
    "tensorflow_serving/apis/predict.proto",
    "tensorflow_serving/apis/classification.proto",
    "tensorflow_serving/apis/input.proto",
    "tensorflow_serving/apis/regression.proto",
    "tensorflow_serving/apis/inference.proto",
    "tensorflow_serving/util/status.proto",
    "tensorflow_serving/apis/get_model_status.proto",
    "tensorflow_serving/apis/model_management.proto",
    "tensorflow_serving/config/log_collector_config.proto",
    "tensorflow_serving/config/logging_config.proto",
    "tensorflow_serving/sources/storage_path/"
    "file_system_storage_path_source.proto",
    "tensorflow_serving/config/model_server_config.proto",
    "tensorflow_serving/apis/model_service.proto",
    "tensorflow_serving/apis/get_model_metadata.proto",
    "tensorflow_serving/apis/prediction_service.proto",
]

REFERENCE_ROOT = "/root/reference/protobuf_srcs"


def build_reference_descriptor_set() -> descriptor_pb2.FileDescriptorSet:
    """Parses the reference closure; adjusts file paths that differ in the
    vendored tree if needed."""
    import os
    rels = []
    for rel in REFERENCE_CLOSURE:
        if os.path.exists(os.path.join(REFERENCE_ROOT, rel)):
            rels.append(rel)
    fdps = parse_files(REFERENCE_ROOT, rels)
    # dependency fix-up: remap `import public` alias paths onto the real
    # file, keep well-known google/protobuf deps (seeded into the pool by
    # the consumer), drop imports outside the parsed closure
    ALIASES = {
        "tensorflow/core/lib/core/error_codes.proto":
            "tensorflow/core/protobuf/error_codes.proto",
    }
    names = {f.name for f in fdps}
    for fdp in fdps:
        deps = list(fdp.dependency)
        del fdp.dependency[:]
        for d in deps:
            d = ALIASES.get(d, d)
            if d in names or d.startswith("google/protobuf/"):
                fdp.dependency.append(d)
    fdset = descriptor_pb2.FileDescriptorSet()
    for fdp in topo_sort(fdps):
        fdset.file.add().CopyFrom(fdp)
    return fdset


def topo_sort(fdps: List[descriptor_pb2.FileDescriptorProto]
              ) -> List[descriptor_pb2.FileDescriptorProto]:
    """Orders files so every dependency precedes its dependents
    (DescriptorPool.Add requires it); unknown deps (well-known types,
    `import public` aliases) are ignored."""
    by_name = {f.name: f for f in fdps}
    out: List[descriptor_pb2.FileDescriptorProto] = []
    done = set()
    visiting = set()

    def visit(name: str):
        if name in done or name not in by_name:
            return
        if name in visiting:
            raise ValueError(f"import cycle at {name}")
        visiting.add(name)
        for dep in by_name[name].dependency:
            visit(dep)
        visiting.discard(name)
        done.add(name)
        out.append(by_name[name])

    for f in fdps:
        visit(f.name)
    return out


def main():
    import sys
    out = sys.argv[1] if len(sys.argv) > 1 else \
        "tests/fixtures/reference_descriptor_set.binpb"
    fdset = build_reference_descriptor_set()
    with open(out, "wb") as fh:
        fh.write(fdset.SerializeToString())
    print(f"wrote {out}: {len(fdset.file)} files, "
          f"{sum(len(f.message_type) for f in fdset.file)} top-level "
          f"messages")


if __name__ == "__main__":
    main()
