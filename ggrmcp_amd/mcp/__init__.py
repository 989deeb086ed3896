"""MCP protocol wire types and validation (reference pkg/mcp)."""

from .types import (  # noqa: F401
    PROTOCOL_VERSION,
    SERVER_NAME,
    SERVER_VERSION,
    AudioContent,
    ContentBlock,
    ImageContent,
    JSONRPCError,
    JSONRPCRequest,
    JSONRPCResponse,
    RPCError,
    TextContent,
    Tool,
    ToolCallResult,
    initialization_result,
    PARSE_ERROR,
    INVALID_REQUEST,
    METHOD_NOT_FOUND,
    INVALID_PARAMS,
    INTERNAL_ERROR,
)
from .validation import ValidationError, Validator, sanitize_error, sanitize_string  # noqa: F401
