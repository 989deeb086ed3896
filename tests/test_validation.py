"""MCP validation tests — mirror reference pkg/mcp/validation.go rules."""

import pytest

from ggrmcp_amd.mcp.validation import (
    ValidationError,
    Validator,
    sanitize_error,
    sanitize_string,
)


@pytest.fixture
def v():
    return Validator()


def test_valid_request(v):
    v.validate_request({"jsonrpc": "2.0", "method": "tools/list", "id": 1})


def test_wrong_jsonrpc_version(v):
    with pytest.raises(ValidationError):
        v.validate_request({"jsonrpc": "1.0", "method": "m", "id": 1})


def test_missing_method(v):
    with pytest.raises(ValidationError):
        v.validate_request({"jsonrpc": "2.0", "id": 1})


def test_method_bad_chars(v):
    with pytest.raises(ValidationError):
        v.validate_request({"jsonrpc": "2.0", "method": "tools list!", "id": 1})


def test_method_too_long(v):
    with pytest.raises(ValidationError):
        v.validate_request({"jsonrpc": "2.0", "method": "a" * 1025, "id": 1})


def test_id_required(v):
    with pytest.raises(ValidationError):
        v.validate_request({"jsonrpc": "2.0", "method": "m"})


def test_id_string_or_number(v):
    v.validate_request({"jsonrpc": "2.0", "method": "m", "id": "abc"})
    v.validate_request({"jsonrpc": "2.0", "method": "m", "id": 3.5})
    with pytest.raises(ValidationError):
        v.validate_request({"jsonrpc": "2.0", "method": "m", "id": [1]})


def test_tool_call_params(v):
    name = v.validate_tool_call_params({"name": "hello_helloservice_sayhello"})
    assert name == "hello_helloservice_sayhello"


def test_tool_call_params_bad_name(v):
    with pytest.raises(ValidationError):
        v.validate_tool_call_params({"name": "bad name!"})
    with pytest.raises(ValidationError):
        v.validate_tool_call_params({"name": "x" * 129})
    with pytest.raises(ValidationError):
        v.validate_tool_call_params({})


def test_arguments_depth_limit(v):
    deep = {"a": 1}
    for _ in range(12):
        deep = {"nest": deep}
    with pytest.raises(ValidationError):
        v.validate_tool_call_params({"name": "t", "arguments": deep})


def test_arguments_string_limit(v):
    with pytest.raises(ValidationError):
        v.validate_tool_call_params({"name": "t", "arguments": {"s": "x" * 1025}})


def test_arguments_size_limit():
    v = Validator(max_args_bytes=100)
    with pytest.raises(ValidationError):
        v.validate_tool_call_params({"name": "t", "arguments": {"s": "y" * 200}})


def test_arguments_ok(v):
    v.validate_tool_call_params(
        {"name": "t", "arguments": {"a": [1, 2, {"b": None, "c": True}], "d": "ok"}}
    )


def test_sanitize_string_strips_control_chars():
    assert sanitize_string("a\x00b\x1fc\nd") == "ab c d".replace(" ", "")[:2] + "c\nd"


def test_sanitize_string_truncates():
    assert len(sanitize_string("x" * 5000)) == 1024


def test_sanitize_error_redacts():
    msg = sanitize_error("failed: password=hunter2 token: abc123 ok=1")
    assert "hunter2" not in msg
    assert "abc123" not in msg
    assert "[REDACTED]" in msg
