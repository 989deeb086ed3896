"""Programmatic protobuf descriptor construction.

There is no ``protoc``/``grpcio-tools`` in this environment, so every proto
schema — the example services mirroring the reference's
``examples/hello-service`` (hello.proto, complex_service.proto), test
fixtures, and the synthetic benchmark services from BASELINE.json — is built
as ``FileDescriptorProto`` objects through these helpers.
"""

from __future__ import annotations

from typing import Dict, Iterable, List, Optional, Sequence, Tuple, Union

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_F = descriptor_pb2.FieldDescriptorProto

# scalar type name -> (proto type enum)
SCALAR_TYPES = {
    "double": _F.TYPE_DOUBLE,
    "float": _F.TYPE_FLOAT,
    "int64": _F.TYPE_INT64,
    "uint64": _F.TYPE_UINT64,
    "int32": _F.TYPE_INT32,
    "fixed64": _F.TYPE_FIXED64,
    "fixed32": _F.TYPE_FIXED32,
    "bool": _F.TYPE_BOOL,
    "string": _F.TYPE_STRING,
    "bytes": _F.TYPE_BYTES,
    "uint32": _F.TYPE_UINT32,
    "sfixed32": _F.TYPE_SFIXED32,
    "sfixed64": _F.TYPE_SFIXED64,
    "sint32": _F.TYPE_SINT32,
    "sint64": _F.TYPE_SINT64,
}


class FileBuilder:
    """Builds one FileDescriptorProto."""

    def __init__(self, name: str, package: str, syntax: str = "proto3") -> None:
        self.fdp = descriptor_pb2.FileDescriptorProto()
        self.fdp.name = name
        self.fdp.package = package
        self.fdp.syntax = syntax
        self._package_prefix = "." + package + "." if package else "."

    def type_ref(self, name: str) -> str:
        """Fully-qualified type reference; pass-through if already qualified."""
        if name.startswith("."):
            return name
        if "." in name and not name[0].isupper():
            return "." + name  # e.g. google.protobuf.Timestamp
        return self._package_prefix + name

    def add_dependency(self, *files: str) -> "FileBuilder":
        for f in files:
            if f not in self.fdp.dependency:
                self.fdp.dependency.append(f)
        return self

    def message(self, name: str) -> "MessageBuilder":
        return MessageBuilder(self, self.fdp.message_type.add(), name)

    def enum(self, name: str, values: Sequence[Tuple[str, int]]) -> "FileBuilder":
        en = self.fdp.enum_type.add()
        en.name = name
        for vname, vnum in values:
            v = en.value.add()
            v.name = vname
            v.number = vnum
        return self

    def service(self, name: str) -> "ServiceBuilder":
        return ServiceBuilder(self, self.fdp.service.add(), name)

    def comment(self, path: Sequence[int], leading: str = "", trailing: str = "") -> "FileBuilder":
        loc = self.fdp.source_code_info.location.add()
        loc.path.extend(path)
        if leading:
            loc.leading_comments = leading
        if trailing:
            loc.trailing_comments = trailing
        return self

    def build(self) -> descriptor_pb2.FileDescriptorProto:
        return self.fdp


class MessageBuilder:
    def __init__(self, file: FileBuilder, proto, name: str) -> None:
        self.file = file
        self.proto = proto
        self.proto.name = name
        self._oneof_indices: Dict[str, int] = {}

    def field(
        self,
        name: str,
        number: int,
        ftype: str,
        repeated: bool = False,
        message: Optional[str] = None,
        enum: Optional[str] = None,
        oneof: Optional[str] = None,
        optional: bool = False,
        json_name: Optional[str] = None,
    ) -> "MessageBuilder":
        f = self.proto.field.add()
        f.name = name
        f.number = number
        f.label = _F.LABEL_REPEATED if repeated else _F.LABEL_OPTIONAL
        if ftype == "message":
            f.type = _F.TYPE_MESSAGE
            f.type_name = self.file.type_ref(message or "")
        elif ftype == "enum":
            f.type = _F.TYPE_ENUM
            f.type_name = self.file.type_ref(enum or "")
        else:
            f.type = SCALAR_TYPES[ftype]
        if json_name:
            f.json_name = json_name
        if oneof is not None:
            if oneof not in self._oneof_indices:
                self._oneof_indices[oneof] = len(self.proto.oneof_decl)
                self.proto.oneof_decl.add().name = oneof
            f.oneof_index = self._oneof_indices[oneof]
        elif optional:
            # proto3 optional = synthetic oneof
            idx = len(self.proto.oneof_decl)
            self.proto.oneof_decl.add().name = "_" + name
            f.oneof_index = idx
            f.proto3_optional = True
        return self

    def map_field(self, name: str, number: int, key_type: str, value_type: str,
                  value_message: Optional[str] = None) -> "MessageBuilder":
        entry_name = "".join(p.capitalize() for p in name.split("_")) + "Entry"
        entry = self.proto.nested_type.add()
        entry.name = entry_name
        entry.options.map_entry = True
        k = entry.field.add()
        k.name = "key"
        k.number = 1
        k.label = _F.LABEL_OPTIONAL
        k.type = SCALAR_TYPES[key_type]
        v = entry.field.add()
        v.name = "value"
        v.number = 2
        v.label = _F.LABEL_OPTIONAL
        if value_type == "message":
            v.type = _F.TYPE_MESSAGE
            v.type_name = self.file.type_ref(value_message or "")
        elif value_type == "enum":
            v.type = _F.TYPE_ENUM
            v.type_name = self.file.type_ref(value_message or "")
        else:
            v.type = SCALAR_TYPES[value_type]
        f = self.proto.field.add()
        f.name = name
        f.number = number
        f.label = _F.LABEL_REPEATED
        f.type = _F.TYPE_MESSAGE
        f.type_name = self.file.type_ref(f"{self.proto.name}.{entry_name}")
        return self

    def done(self) -> FileBuilder:
        return self.file


class ServiceBuilder:
    def __init__(self, file: FileBuilder, proto, name: str) -> None:
        self.file = file
        self.proto = proto
        self.proto.name = name

    def method(
        self,
        name: str,
        input_type: str,
        output_type: str,
        client_streaming: bool = False,
        server_streaming: bool = False,
    ) -> "ServiceBuilder":
        m = self.proto.method.add()
        m.name = name
        m.input_type = self.file.type_ref(input_type)
        m.output_type = self.file.type_ref(output_type)
        m.client_streaming = client_streaming
        m.server_streaming = server_streaming
        return self

    def done(self) -> FileBuilder:
        return self.file


def pool_for(fdps: Iterable[descriptor_pb2.FileDescriptorProto]) -> descriptor_pool.DescriptorPool:
    from ..descriptors.loader import build_pool

    return build_pool(fdps)


def message_class(pool: descriptor_pool.DescriptorPool, full_name: str):
    return message_factory.GetMessageClass(pool.FindMessageTypeByName(full_name))
