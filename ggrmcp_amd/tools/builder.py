"""protobuf MessageDescriptor -> JSON Schema tool builder.

Re-design of the reference's ``pkg/tools/builder.go``:

* one MCP tool per unary method with input+output schema and validation
  (builder.go:36-89); description fallback "Calls the X method of the Y
  service" (builder.go:92-100);
* recursive message schemas with cycle detection via a visited set ->
  ``$ref: #/definitions/<fqn>`` (builder.go:162-174);
* required = proto3 fields without presence (no ``optional``, not in a
  oneof) (builder.go:206-211);
* oneofs -> JSON-Schema ``oneOf`` of single-required-field objects
  (builder.go:214-253);
* repeated -> array (builder.go:272-281); maps ->
  ``patternProperties: {".*": valueSchema}`` (builder.go:284-297);
* scalar table incl. 64-bit ints as ``{"type":"integer","format":"int64"}``
  (builder.go:307-342); enums as string with values + enumDescriptions
  (builder.go:344-371);
* 13 well-known types special-cased (builder.go:373-427).

Two deliberate improvements over the reference: the schema cache actually
works (the reference declares ``schemaCache`` and never populates it,
builder.go:18,29 — schemas there are rebuilt on every tools/list), and
server-streaming methods get tools too (the reference skips all streaming,
builder.go:129-135; this gateway supports server-streaming calls).
"""

from __future__ import annotations

import logging
from typing import Any, Dict, Iterable, List, Optional, Tuple

from google.protobuf import descriptor_pb2
from google.protobuf.descriptor import Descriptor, FieldDescriptor

from ..mcp.types import Tool
from ..types import MethodInfo

log = logging.getLogger("ggrmcp.tools")

_F = FieldDescriptor

# scalar proto type -> JSON schema fragment (builder.go:307-342)
_SCALAR_SCHEMAS: Dict[int, Dict[str, Any]] = {
    _F.TYPE_DOUBLE: {"type": "number", "format": "double"},
    _F.TYPE_FLOAT: {"type": "number", "format": "float"},
    _F.TYPE_INT64: {"type": "integer", "format": "int64"},
    _F.TYPE_UINT64: {"type": "integer", "format": "uint64"},
    _F.TYPE_INT32: {"type": "integer", "format": "int32"},
    _F.TYPE_FIXED64: {"type": "integer", "format": "uint64"},
    _F.TYPE_FIXED32: {"type": "integer", "format": "uint32"},
    _F.TYPE_BOOL: {"type": "boolean"},
    _F.TYPE_STRING: {"type": "string"},
    _F.TYPE_BYTES: {"type": "string", "format": "byte"},
    _F.TYPE_UINT32: {"type": "integer", "format": "uint32"},
    _F.TYPE_SFIXED32: {"type": "integer", "format": "int32"},
    _F.TYPE_SFIXED64: {"type": "integer", "format": "int64"},
    _F.TYPE_SINT32: {"type": "integer", "format": "int32"},
    _F.TYPE_SINT64: {"type": "integer", "format": "int64"},
}

# the 13 well-known types (builder.go:373-427)
_WKT_SCHEMAS: Dict[str, Dict[str, Any]] = {
    "google.protobuf.Any": {
        "type": "object",
        "properties": {"@type": {"type": "string"}},
        "additionalProperties": True,
        "description": "google.protobuf.Any: arbitrary message with @type URL",
    },
    "google.protobuf.Timestamp": {"type": "string", "format": "date-time"},
    "google.protobuf.Duration": {
        "type": "string",
        "pattern": r"^-?\d+(\.\d+)?s$",
        "description": "Duration in seconds, e.g. '3.5s'",
    },
    "google.protobuf.Struct": {"type": "object", "additionalProperties": True},
    "google.protobuf.Value": {
        "description": "Any JSON value (google.protobuf.Value)"
    },
    "google.protobuf.ListValue": {"type": "array"},
    "google.protobuf.Empty": {"type": "object", "additionalProperties": False},
    "google.protobuf.FieldMask": {"type": "string"},
    "google.protobuf.DoubleValue": {"type": ["number", "null"]},
    "google.protobuf.FloatValue": {"type": ["number", "null"]},
    "google.protobuf.Int64Value": {"type": ["integer", "null"], "format": "int64"},
    "google.protobuf.UInt64Value": {"type": ["integer", "null"], "format": "uint64"},
    "google.protobuf.Int32Value": {"type": ["integer", "null"], "format": "int32"},
    "google.protobuf.UInt32Value": {"type": ["integer", "null"], "format": "uint32"},
    "google.protobuf.BoolValue": {"type": ["boolean", "null"]},
    "google.protobuf.StringValue": {"type": ["string", "null"]},
    "google.protobuf.BytesValue": {"type": ["string", "null"], "format": "byte"},
}


def build_comment_index(
    fdps: Iterable[descriptor_pb2.FileDescriptorProto],
) -> Dict[str, str]:
    """full proto name (message or message.field) -> leading/trailing comment.

    The reference pulls comments through protoreflect's SourceLocations
    (builder.go:441-462); Python descriptors carry no source info, so the
    index is built from the FileDescriptorProtos directly using the standard
    SourceCodeInfo path convention.
    """
    out: Dict[str, str] = {}
    for fdp in fdps:
        locs = {
            tuple(loc.path): (loc.leading_comments or loc.trailing_comments).strip()
            for loc in fdp.source_code_info.location
            if loc.leading_comments or loc.trailing_comments
        }
        if not locs:
            continue
        prefix = fdp.package + "." if fdp.package else ""

        def walk_message(msg, path: Tuple[int, ...], scope: str) -> None:
            fq = scope + msg.name
            c = locs.get(path)
            if c:
                out[fq] = c
            for fi, fld in enumerate(msg.field):
                c = locs.get(path + (2, fi))
                if c:
                    out[f"{fq}.{fld.name}"] = c
            for ni, nested in enumerate(msg.nested_type):
                walk_message(nested, path + (3, ni), fq + ".")

        for mi, msg in enumerate(fdp.message_type):
            walk_message(msg, (4, mi), prefix)
        for si, svc in enumerate(fdp.service):
            c = locs.get((6, si))
            if c:
                out[prefix + svc.name] = c
            for mj, method in enumerate(svc.method):
                c = locs.get((6, si, 2, mj))
                if c:
                    out[f"{prefix}{svc.name}.{method.name}"] = c
    return out


class MCPToolBuilder:
    """Reference MCPToolBuilder (builder.go:26-122)."""

    def __init__(self, comment_index: Optional[Dict[str, str]] = None) -> None:
        self.comments = comment_index or {}
        # working cache: message full name -> built schema (per tools version)
        self._schema_cache: Dict[str, Dict[str, Any]] = {}
        self._cache_key: Any = None

    def set_cache_key(self, key: Any) -> None:
        """Invalidate the schema cache when the tool map version changes."""
        if key != self._cache_key:
            self._schema_cache.clear()
            self._cache_key = key

    # -- tools (builder.go:36-135) -------------------------------------------

    def build_tools(self, methods: Iterable[MethodInfo]) -> List[Tool]:
        tools = []
        for mi in methods:
            if mi.is_client_streaming:
                # reference skips all streaming (builder.go:129-135); we lift
                # the restriction for server-streaming only.
                continue
            try:
                tools.append(self.build_tool(mi))
            except Exception as e:
                log.warning("failed to build tool for %s: %s", mi.full_method_path, e)
        return tools

    def build_tool(self, mi: MethodInfo) -> Tool:
        description = mi.description or self.comments.get(
            f"{mi.full_service_name or mi.service_name}.{mi.method_name}", ""
        )
        if not description:
            # builder.go:92-100 fallback
            description = f"Calls the {mi.method_name} method of the {mi.service_name} service"
        if mi.is_server_streaming:
            description += " (server-streaming)"
        tool = Tool(
            name=mi.tool_name(),
            description=description,
            input_schema=self.extract_message_schema(mi.input_descriptor),
            output_schema=self.extract_message_schema(mi.output_descriptor),
        )
        self._validate_tool(tool)
        return tool

    def _validate_tool(self, tool: Tool) -> None:
        """builder.go:103-122."""
        if not tool.name:
            raise ValueError("tool has no name")
        if not isinstance(tool.input_schema, dict) or not tool.input_schema:
            raise ValueError(f"tool {tool.name} has no input schema")

    # -- schemas (builder.go:156-434) -----------------------------------------

    def extract_message_schema(self, desc: Descriptor) -> Dict[str, Any]:
        """Root entry: schema + collected $ref definitions (builder.go:156-174)."""
        cached = self._schema_cache.get(desc.full_name)
        if cached is not None:
            return cached
        definitions: Dict[str, Dict[str, Any]] = {}
        schema = self._message_schema(desc, visited=set(), definitions=definitions)
        if definitions:
            schema = dict(schema)
            schema["definitions"] = definitions
        self._schema_cache[desc.full_name] = schema
        return schema

    def _message_schema(
        self,
        desc: Descriptor,
        visited: set,
        definitions: Dict[str, Dict[str, Any]],
    ) -> Dict[str, Any]:
        wkt = _WKT_SCHEMAS.get(desc.full_name)
        if wkt is not None:
            return dict(wkt)
        if desc.full_name in visited:
            # circular reference -> $ref (builder.go:162-174)
            if desc.full_name not in definitions:
                definitions[desc.full_name] = {}  # placeholder; filled by owner
            return {"$ref": f"#/definitions/{desc.full_name}"}
        visited = visited | {desc.full_name}

        properties: Dict[str, Any] = {}
        required: List[str] = []
        # real (non-synthetic) oneofs -> oneOf constraint (builder.go:214-253)
        oneof_groups: Dict[str, List[str]] = {}

        for field in desc.fields:
            json_name = field.json_name or field.name
            fschema = self._field_schema(field, visited, definitions)
            comment = self.comments.get(f"{desc.full_name}.{field.name}")
            if comment:
                fschema = dict(fschema)
                fschema.setdefault("description", comment)
            properties[json_name] = fschema
            oneof = field.containing_oneof
            if oneof is not None:
                if not _is_synthetic_oneof(field):
                    oneof_groups.setdefault(oneof.name, []).append(json_name)
            elif not field.is_repeated and not field.has_presence:
                required.append(json_name)

        schema: Dict[str, Any] = {"type": "object", "properties": properties}
        title = desc.full_name
        schema["title"] = title
        comment = self.comments.get(desc.full_name)
        if comment:
            schema["description"] = comment
        if required:
            schema["required"] = sorted(required)
        if oneof_groups:
            one_ofs = []
            for group, names in sorted(oneof_groups.items()):
                one_ofs.append(
                    {
                        "oneOf": [{"required": [n]} for n in sorted(names)]
                        + [{"not": {"anyOf": [{"required": [n]} for n in sorted(names)]}}]
                    }
                )
            schema["allOf"] = one_ofs
        schema["additionalProperties"] = False

        # if something $ref'd us, fill the definition (cycle owner)
        if desc.full_name in definitions and not definitions[desc.full_name]:
            definitions[desc.full_name] = {
                k: v for k, v in schema.items() if k != "definitions"
            }
        return schema

    def _field_schema(
        self,
        field: FieldDescriptor,
        visited: set,
        definitions: Dict[str, Dict[str, Any]],
    ) -> Dict[str, Any]:
        # maps (builder.go:284-297)
        if _is_map_field(field):
            value_field = field.message_type.fields_by_name["value"]
            value_schema = self._field_schema(value_field, visited, definitions)
            return {
                "type": "object",
                "patternProperties": {".*": value_schema},
                "additionalProperties": False,
            }
        base = self._singular_schema(field, visited, definitions)
        if field.is_repeated:
            return {"type": "array", "items": base}
        return base

    def _singular_schema(
        self,
        field: FieldDescriptor,
        visited: set,
        definitions: Dict[str, Dict[str, Any]],
    ) -> Dict[str, Any]:
        if field.type == _F.TYPE_MESSAGE:
            return self._message_schema(field.message_type, visited, definitions)
        if field.type == _F.TYPE_ENUM:
            ed = field.enum_type
            values = [v.name for v in ed.values]
            # protojson also accepts enum numbers (builder.go:344-371 emits
            # names; numbers documented via enumDescriptions)
            return {
                "type": "string",
                "enum": values,
                "enumDescriptions": {v.name: str(v.number) for v in ed.values},
                "title": ed.full_name,
            }
        if field.type == _F.TYPE_GROUP:
            raise ValueError("proto2 groups are not supported")
        return dict(_SCALAR_SCHEMAS[field.type])


def _is_map_field(field: FieldDescriptor) -> bool:
    return (
        field.type == _F.TYPE_MESSAGE
        and field.message_type.GetOptions().map_entry
    )


def _is_synthetic_oneof(field: FieldDescriptor) -> bool:
    oneof = field.containing_oneof
    return oneof is not None and len(oneof.fields) == 1 and oneof.name == "_" + field.name


