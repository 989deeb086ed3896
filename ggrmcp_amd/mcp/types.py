"""MCP / JSON-RPC 2.0 wire types.

Re-design of the reference's ``pkg/mcp/types.go``: JSON-RPC request/response/
error structs (types.go:41-75), RequestID string-or-number semantics
(types.go:9-38), standard error codes (types.go:69-75), initialize result +
capabilities (types.go:89-116), content blocks (types.go:119-159), Tool and
ToolCallResult (types.go:162-173).

These are plain dataclasses with ``to_dict`` producing exactly the JSON wire
shape; the GPU response-assembly kernel emits the same shapes byte-for-byte
(differentially tested against this module).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Union

# JSON-RPC 2.0 standard error codes (reference types.go:69-75).
PARSE_ERROR = -32700
INVALID_REQUEST = -32600
METHOD_NOT_FOUND = -32601
INVALID_PARAMS = -32602
INTERNAL_ERROR = -32603

PROTOCOL_VERSION = "2024-11-05"  # reference handler.go:160-179
SERVER_NAME = "ggrmcp-amd"
SERVER_VERSION = "1.0.0"

# RequestID: JSON-RPC ids may be a string or a number (types.go:9-38).
RequestID = Union[str, int, float, None]


@dataclass
class RPCError:
    code: int
    message: str
    data: Any = None

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"code": self.code, "message": self.message}
        if self.data is not None:
            d["data"] = self.data
        return d


class JSONRPCError(Exception):
    """Exception carrying a JSON-RPC error (maps to RPCError on the wire)."""

    def __init__(self, code: int, message: str, data: Any = None) -> None:
        super().__init__(message)
        self.error = RPCError(code, message, data)


@dataclass
class JSONRPCRequest:
    jsonrpc: str = "2.0"
    method: str = ""
    params: Optional[Dict[str, Any]] = None
    id: RequestID = None
    has_id: bool = False  # JSON null id vs absent id distinction

    @classmethod
    def from_dict(cls, data: Dict[str, Any]) -> "JSONRPCRequest":
        return cls(
            jsonrpc=data.get("jsonrpc", ""),
            method=data.get("method", ""),
            params=data.get("params"),
            id=data.get("id"),
            has_id="id" in data,
        )

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"jsonrpc": self.jsonrpc, "method": self.method}
        if self.params is not None:
            d["params"] = self.params
        if self.has_id or self.id is not None:
            d["id"] = self.id
        return d


@dataclass
class JSONRPCResponse:
    id: RequestID = None
    result: Any = None
    error: Optional[RPCError] = None
    jsonrpc: str = "2.0"

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"jsonrpc": self.jsonrpc, "id": self.id}
        if self.error is not None:
            d["error"] = self.error.to_dict()
        else:
            d["result"] = self.result
        return d


# ---- content blocks (reference types.go:119-159) ---------------------------


@dataclass
class ContentBlock:
    type: str
    data: Dict[str, Any] = field(default_factory=dict)

    def to_dict(self) -> Dict[str, Any]:
        return {"type": self.type, **self.data}


def TextContent(text: str) -> ContentBlock:
    return ContentBlock("text", {"text": text})


def ImageContent(data_b64: str, mime_type: str) -> ContentBlock:
    return ContentBlock("image", {"data": data_b64, "mimeType": mime_type})


def AudioContent(data_b64: str, mime_type: str) -> ContentBlock:
    return ContentBlock("audio", {"data": data_b64, "mimeType": mime_type})


# ---- tools (reference types.go:162-173) ------------------------------------


@dataclass
class Tool:
    name: str
    description: str = ""
    input_schema: Dict[str, Any] = field(default_factory=dict)
    output_schema: Optional[Dict[str, Any]] = None

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {
            "name": self.name,
            "description": self.description,
            "inputSchema": self.input_schema,
        }
        if self.output_schema is not None:
            d["outputSchema"] = self.output_schema
        return d


@dataclass
class ToolCallResult:
    content: List[ContentBlock] = field(default_factory=list)
    is_error: bool = False

    def to_dict(self) -> Dict[str, Any]:
        return {
            "content": [c.to_dict() for c in self.content],
            "isError": self.is_error,
        }


def initialization_result() -> Dict[str, Any]:
    """Reference handler.go:160-179: fixed initialize payload."""
    return {
        "protocolVersion": PROTOCOL_VERSION,
        "capabilities": {
            "tools": {"listChanged": False},
            "prompts": {"listChanged": False},
            "resources": {"listChanged": False, "subscribe": False},
        },
        "serverInfo": {"name": SERVER_NAME, "version": SERVER_VERSION},
    }
