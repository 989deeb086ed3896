"""JSON-RPC / tool-call request validation and sanitization.

Re-design of the reference's ``pkg/mcp/validation.go``:

* request envelope rules (validation.go:24-61): jsonrpc must be "2.0",
  method non-empty, <= 1024 chars, matching ``^[a-zA-Z0-9_/]+$``, id required;
* tools/call params (validation.go:96-125): ``name`` is a string <= 128 chars
  matching ``^[a-zA-Z0-9_\\.]+$``;
* recursive argument limits (validation.go:163-218): nesting depth <= 10,
  approximate encoded size <= 1 MB, strings <= 1024 chars;
* ``sanitize_string`` strips control characters and truncates to 1024
  (validation.go:235-245);
* ``sanitize_error`` redacts secret-looking tokens (validation.go:248-271).

The same limits are compiled into the GPU validator kernel's automaton
(ggrmcp_amd/ops/csrc/): this module is the CPU oracle it is differentially
tested against.
"""

from __future__ import annotations

import re
from typing import Any, Dict, List, Optional

MAX_METHOD_LEN = 1024
MAX_TOOL_NAME_LEN = 128
MAX_STRING_LEN = 1024
MAX_DEPTH = 10
MAX_ARGS_BYTES = 1024 * 1024

_METHOD_RE = re.compile(r"^[a-zA-Z0-9_/]+$")
_TOOL_NAME_RE = re.compile(r"^[a-zA-Z0-9_\.]+$")
# reference validation.go:248-271 secret-token redaction
_SECRET_RE = re.compile(
    r"(?i)(password|token|key|secret|credential|auth)[a-z0-9_\-]*\s*[=:]\s*\S+"
)
_CONTROL_RE = re.compile(r"[\x00-\x08\x0b\x0c\x0e-\x1f\x7f]")


class ValidationError(ValueError):
    def __init__(self, field: str, message: str) -> None:
        super().__init__(f"{field}: {message}")
        self.field = field
        self.message = message


class Validator:
    """Reference pkg/mcp/validation.go Validator."""

    def __init__(
        self,
        max_depth: int = MAX_DEPTH,
        max_args_bytes: int = MAX_ARGS_BYTES,
        max_string_len: int = MAX_STRING_LEN,
    ) -> None:
        self.max_depth = max_depth
        self.max_args_bytes = max_args_bytes
        self.max_string_len = max_string_len

    # -- envelope (validation.go:24-61) -------------------------------------

    def validate_request(self, data: Dict[str, Any]) -> None:
        if not isinstance(data, dict):
            raise ValidationError("request", "request body must be a JSON object")
        if data.get("jsonrpc") != "2.0":
            raise ValidationError("jsonrpc", 'must be "2.0"')
        method = data.get("method")
        if not isinstance(method, str) or not method:
            raise ValidationError("method", "required and must be a non-empty string")
        if len(method) > MAX_METHOD_LEN:
            raise ValidationError("method", f"exceeds {MAX_METHOD_LEN} characters")
        if not _METHOD_RE.match(method):
            raise ValidationError("method", "contains invalid characters")
        if "id" not in data:
            raise ValidationError("id", "required")
        rid = data["id"]
        if rid is not None and not isinstance(rid, (str, int, float)):
            raise ValidationError("id", "must be a string or number")
        params = data.get("params")
        if params is not None and not isinstance(params, dict):
            raise ValidationError("params", "must be an object")

    # -- tools/call params (validation.go:96-125) ----------------------------

    def validate_tool_call_params(self, params: Optional[Dict[str, Any]]) -> str:
        if not isinstance(params, dict):
            raise ValidationError("params", "required for tools/call")
        name = params.get("name")
        if not isinstance(name, str) or not name:
            raise ValidationError("name", "required and must be a non-empty string")
        if len(name) > MAX_TOOL_NAME_LEN:
            raise ValidationError("name", f"exceeds {MAX_TOOL_NAME_LEN} characters")
        if not _TOOL_NAME_RE.match(name):
            raise ValidationError("name", "contains invalid characters")
        args = params.get("arguments")
        if args is not None:
            self.validate_arguments(args)
        return name

    # -- recursive argument limits (validation.go:163-218) -------------------

    def validate_arguments(self, args: Any) -> None:
        size = self._validate_value(args, 0, "arguments")
        if size > self.max_args_bytes:
            raise ValidationError(
                "arguments", f"approximate size {size} exceeds {self.max_args_bytes}"
            )

    def _validate_value(self, value: Any, depth: int, path: str) -> int:
        if depth > self.max_depth:
            raise ValidationError(path, f"nesting depth exceeds {self.max_depth}")
        if value is None:
            return 4
        if isinstance(value, bool):
            return 5
        if isinstance(value, (int, float)):
            return 20
        if isinstance(value, str):
            if len(value) > self.max_string_len:
                raise ValidationError(path, f"string exceeds {self.max_string_len} characters")
            return len(value) + 2
        if isinstance(value, list):
            return 2 + sum(
                self._validate_value(v, depth + 1, f"{path}[{i}]") for i, v in enumerate(value)
            )
        if isinstance(value, dict):
            total = 2
            for k, v in value.items():
                if not isinstance(k, str):
                    raise ValidationError(path, "object keys must be strings")
                if len(k) > self.max_string_len:
                    raise ValidationError(path, f"key exceeds {self.max_string_len} characters")
                total += len(k) + 4 + self._validate_value(v, depth + 1, f"{path}.{k}")
            return total
        raise ValidationError(path, f"unsupported value type {type(value).__name__}")


def sanitize_string(s: str, max_len: int = MAX_STRING_LEN) -> str:
    """Strip control chars + truncate (reference validation.go:235-245)."""
    s = _CONTROL_RE.sub("", s)
    if len(s) > max_len:
        s = s[:max_len]
    return s


def sanitize_error(message: str) -> str:
    """Redact secret-looking tokens (reference validation.go:248-271)."""
    return sanitize_string(_SECRET_RE.sub(r"\1=[REDACTED]", message))
