"""MCP tool building: protobuf descriptors -> JSON Schema (reference pkg/tools)."""

from .builder import MCPToolBuilder, build_comment_index  # noqa: F401
