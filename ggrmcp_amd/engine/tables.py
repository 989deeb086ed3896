"""Descriptor -> flat GPU transcode tables.

The reference resolves protobuf structure reflectively on every call
(dynamicpb.NewMessage + protojson, reflection.go:351-381).  Here the
structure is compiled ONCE per tool-map version into flat, GPU-resident
tables; the HIP kernels walk these instead of doing any reflection:

* ``msg_table``   — per message type: field range + well-known-type kind;
* ``field_table`` — per field: FNV-1a hashes of both accepted JSON keys
  (json_name and proto name), field number, wire kind, flags, sub-message /
  enum index, name offsets into the blob;  sorted by field number inside
  each message (the decode kernel scans by number, the encode kernel
  compares hashes);
* ``enum_table`` / ``enum_values`` — value name hash <-> number, both
  directions;
* ``tool_table``  — tool-name hash -> input/output message indices (the GPU
  resolves ``params.name`` itself during envelope parsing);
* ``name_blob``   — UTF-8 bytes of every name referenced above.

Layouts are mirrored by static_asserts in ops/csrc/common.h; keep in sync.
Tables are deterministic functions of the (sorted) tool map, so every DP
rank compiles identical bytes from the broadcast descriptor blob
(parallel/dist.py) — verified by checksum in tests.
"""

from __future__ import annotations

import struct
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from google.protobuf.descriptor import Descriptor, EnumDescriptor, FieldDescriptor

from ..types import MethodInfo

_F = FieldDescriptor

# ---- binary record formats (little endian) ---------------------------------

FIELD_ENTRY_FMT = "<QQIIIHHiBBBB"  # 40 bytes
FIELD_ENTRY_SIZE = struct.calcsize(FIELD_ENTRY_FMT)
MSG_ENTRY_FMT = "<iiii"  # field_start, field_count, wkt_kind, flags
MSG_ENTRY_SIZE = struct.calcsize(MSG_ENTRY_FMT)
ENUM_ENTRY_FMT = "<ii"  # val_start, val_count
ENUM_VALUE_FMT = "<QiIHHi"  # hash, number, name_off, name_len, pad, pad2 -> 24 B
# (24 keeps the u64 hash 8-byte aligned across array elements)
ENUM_VALUE_SIZE = struct.calcsize(ENUM_VALUE_FMT)
TOOL_ENTRY_FMT = "<QiiIHH"  # hash, in_msg, out_msg, name_off, name_len, flags
TOOL_ENTRY_SIZE = struct.calcsize(TOOL_ENTRY_FMT)

assert FIELD_ENTRY_SIZE == 40, FIELD_ENTRY_SIZE
assert ENUM_VALUE_SIZE == 24
assert TOOL_ENTRY_SIZE == 24

# field flags
F_REPEATED = 1
F_PACKED = 2
F_MAP = 4
F_HAS_PRESENCE = 8
F_ONEOF = 16

# msg wkt kinds (msg_table.wkt_kind)
WKT_NONE = 0
WKT_TIMESTAMP = 1
WKT_DURATION = 2
WKT_STRUCT = 3
WKT_VALUE = 4
WKT_LISTVALUE = 5
WKT_ANY = 6
WKT_FIELDMASK = 7
WKT_EMPTY = 8
WKT_WRAPPER = 9  # value schema described by its single field entry

_WKT_BY_NAME = {
    "google.protobuf.Timestamp": WKT_TIMESTAMP,
    "google.protobuf.Duration": WKT_DURATION,
    "google.protobuf.Struct": WKT_STRUCT,
    "google.protobuf.Value": WKT_VALUE,
    "google.protobuf.ListValue": WKT_LISTVALUE,
    "google.protobuf.Any": WKT_ANY,
    "google.protobuf.FieldMask": WKT_FIELDMASK,
    "google.protobuf.Empty": WKT_EMPTY,
    "google.protobuf.DoubleValue": WKT_WRAPPER,
    "google.protobuf.FloatValue": WKT_WRAPPER,
    "google.protobuf.Int64Value": WKT_WRAPPER,
    "google.protobuf.UInt64Value": WKT_WRAPPER,
    "google.protobuf.Int32Value": WKT_WRAPPER,
    "google.protobuf.UInt32Value": WKT_WRAPPER,
    "google.protobuf.BoolValue": WKT_WRAPPER,
    "google.protobuf.StringValue": WKT_WRAPPER,
    "google.protobuf.BytesValue": WKT_WRAPPER,
}

# tool flags
T_SERVER_STREAMING = 1


def fnv1a64(data: bytes) -> int:
    h = 0xCBF29CE484222325
    for b in data:
        h ^= b
        h = (h * 0x100000001B3) & 0xFFFFFFFFFFFFFFFF
    return h


@dataclass
class CompiledTables:
    """Host-side image of the GPU tables."""

    msg_table: bytes
    field_table: bytes
    enum_table: bytes
    enum_values: bytes
    tool_table: bytes
    name_blob: bytes
    n_msgs: int
    n_tools: int
    msg_index: Dict[str, int]  # message full name -> index
    tool_index: Dict[str, int]  # tool name -> index
    tool_order: List[str]  # index -> tool name

    def blobs(self) -> Tuple[bytes, ...]:
        return (
            self.msg_table,
            self.field_table,
            self.enum_table,
            self.enum_values,
            self.tool_table,
            self.name_blob,
        )

    def checksum(self) -> int:
        h = 0xCBF29CE484222325
        for blob in self.blobs():
            h ^= fnv1a64(blob)
        return h


class TableCompiler:
    def __init__(self) -> None:
        self.msg_index: Dict[str, int] = {}
        self.enum_index: Dict[str, int] = {}
        self.msgs: List[Descriptor] = []
        self.enums: List[EnumDescriptor] = []
        self.blob = bytearray()
        self._blob_cache: Dict[bytes, int] = {}

    def _intern(self, s: str) -> Tuple[int, int]:
        data = s.encode("utf-8")
        off = self._blob_cache.get(data)
        if off is None:
            off = len(self.blob)
            self.blob.extend(data)
            self._blob_cache[data] = off
        return off, len(data)

    def _add_message(self, desc: Descriptor) -> int:
        idx = self.msg_index.get(desc.full_name)
        if idx is not None:
            return idx
        idx = len(self.msgs)
        self.msg_index[desc.full_name] = idx
        self.msgs.append(desc)
        # recurse into field types
        for f in desc.fields:
            if f.type == _F.TYPE_MESSAGE:
                self._add_message(f.message_type)
            elif f.type == _F.TYPE_ENUM:
                self._add_enum(f.enum_type)
        return idx

    def _add_enum(self, desc: EnumDescriptor) -> int:
        idx = self.enum_index.get(desc.full_name)
        if idx is not None:
            return idx
        idx = len(self.enums)
        self.enum_index[desc.full_name] = idx
        self.enums.append(desc)
        return idx

    def compile(self, tools: Dict[str, MethodInfo]) -> CompiledTables:
        # deterministic ordering: sorted tool names drive everything
        tool_names = sorted(tools)
        tool_rows: List[Tuple[str, int, int, int]] = []
        for name in tool_names:
            mi = tools[name]
            in_idx = self._add_message(mi.input_descriptor)
            out_idx = self._add_message(mi.output_descriptor)
            flags = T_SERVER_STREAMING if mi.is_server_streaming else 0
            tool_rows.append((name, in_idx, out_idx, flags))

        # message + field tables (self.msgs grows during iteration as nested
        # types are discovered, so iterate by index)
        msg_entries = bytearray()
        field_entries = bytearray()
        i = 0
        while i < len(self.msgs):
            desc = self.msgs[i]
            i += 1
            wkt = _WKT_BY_NAME.get(desc.full_name, WKT_NONE)
            fields = sorted(desc.fields, key=lambda f: f.number)
            field_start = len(field_entries) // FIELD_ENTRY_SIZE
            for f in fields:
                flags = 0
                if f.is_repeated:
                    flags |= F_REPEATED
                if f.is_packed:
                    flags |= F_PACKED
                is_map = (
                    f.type == _F.TYPE_MESSAGE and f.message_type.GetOptions().map_entry
                )
                if is_map:
                    flags |= F_MAP
                if f.has_presence:
                    flags |= F_HAS_PRESENCE
                oneof_id = 255
                if f.containing_oneof is not None and not (
                    len(f.containing_oneof.fields) == 1
                    and f.containing_oneof.name == "_" + f.name
                ):
                    flags |= F_ONEOF
                    oneof_id = f.containing_oneof.index
                sub = -1
                if f.type == _F.TYPE_MESSAGE:
                    sub = self._add_message(f.message_type)
                elif f.type == _F.TYPE_ENUM:
                    sub = self._add_enum(f.enum_type)
                name_off, name_len = self._intern(f.name)
                json_name = f.json_name or f.camelcase_name or f.name
                json_off, json_len = self._intern(json_name)
                field_entries.extend(
                    struct.pack(
                        FIELD_ENTRY_FMT,
                        fnv1a64(json_name.encode()),
                        fnv1a64(f.name.encode()),
                        f.number,
                        name_off,
                        json_off,
                        name_len,
                        json_len,
                        sub,
                        f.type,
                        flags,
                        oneof_id,
                        0,
                    )
                )
            msg_entries.extend(
                struct.pack(
                    MSG_ENTRY_FMT, field_start, len(fields), wkt, 0
                )
            )

        # enum tables
        enum_entries = bytearray()
        enum_value_entries = bytearray()
        for desc in self.enums:
            vals = sorted(desc.values, key=lambda v: v.number)
            start = len(enum_value_entries) // ENUM_VALUE_SIZE
            for v in vals:
                name_off, name_len = self._intern(v.name)
                enum_value_entries.extend(
                    struct.pack(
                        ENUM_VALUE_FMT,
                        fnv1a64(v.name.encode()),
                        v.number,
                        name_off,
                        name_len,
                        0,
                        0,
                    )
                )
            enum_entries.extend(struct.pack(ENUM_ENTRY_FMT, start, len(vals)))

        # tool table
        tool_entries = bytearray()
        for name, in_idx, out_idx, flags in tool_rows:
            name_off, name_len = self._intern(name)
            tool_entries.extend(
                struct.pack(
                    TOOL_ENTRY_FMT,
                    fnv1a64(name.encode()),
                    in_idx,
                    out_idx,
                    name_off,
                    name_len,
                    flags,
                )
            )

        return CompiledTables(
            msg_table=bytes(msg_entries),
            field_table=bytes(field_entries),
            enum_table=bytes(enum_entries),
            enum_values=bytes(enum_value_entries),
            tool_table=bytes(tool_entries),
            name_blob=bytes(self.blob),
            n_msgs=len(self.msgs),
            n_tools=len(tool_rows),
            msg_index=dict(self.msg_index),
            tool_index={name: i for i, (name, *_r) in enumerate(tool_rows)},
            tool_order=[name for name, *_r in tool_rows],
        )


def compile_tables(tools: Dict[str, MethodInfo]) -> CompiledTables:
    return TableCompiler().compile(tools)
