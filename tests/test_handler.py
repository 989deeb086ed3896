"""MCP handler tests over the full middleware chain against a live in-process
backend — mirrors the reference's integration_test.go JSON-RPC workflow tests
(tests/integration_test.go:196-298), error paths (:300-410), session headers
(:433-483), and handler_header_test.go filtering scenarios."""

import asyncio
import json

import pytest

from examples.hello_service import serve
from ggrmcp_amd.config import Config
from ggrmcp_amd.backend.discovery import ServiceDiscoverer
from ggrmcp_amd.server.handler import MCPHandler
from ggrmcp_amd.server.middleware import (
    MetricsRecorder,
    Request,
    chain_middleware,
    default_middleware,
)
from ggrmcp_amd.tools import MCPToolBuilder, build_comment_index


@pytest.fixture(scope="module")
def env():
    server, target = serve("127.0.0.1:0")
    cfg = Config.default()
    host, _, port = target.rpartition(":")
    cfg.grpc.host, cfg.grpc.port = host, int(port)
    cfg.server.rate_limit_rps = 100000.0  # don't throttle tests
    cfg.server.rate_limit_burst = 100000
    d = ServiceDiscoverer(cfg)
    d.connect(timeout_s=10)
    d.discover()
    comment_index = build_comment_index(f for b in d._fdps for f in b)
    handler = MCPHandler(d, tool_builder=MCPToolBuilder(comment_index), config=cfg)
    recorder = MetricsRecorder()
    chained = chain_middleware(handler.handle, default_middleware(cfg.server, recorder))
    yield handler, chained, recorder
    d.close()
    server.stop(grace=None)


def post(chained, payload, headers=None):
    req = Request(
        method="POST",
        path="/",
        headers={"content-type": "application/json", **(headers or {})},
        body=json.dumps(payload).encode(),
    )
    resp = asyncio.run(chained(req))
    body = json.loads(resp.body) if resp.body else None
    return resp, body


def test_initialize_get(env):
    _, chained, _ = env
    resp = asyncio.run(chained(Request(method="GET", path="/")))
    assert resp.status == 200
    assert "Mcp-Session-Id" in resp.headers
    body = json.loads(resp.body)
    assert body["result"]["protocolVersion"] == "2024-11-05"
    # security + CORS headers applied by middleware
    assert resp.headers["X-Frame-Options"] == "DENY"
    assert resp.headers["Access-Control-Expose-Headers"] == "Mcp-Session-Id"


def test_initialize_post(env):
    _, chained, _ = env
    resp, body = post(chained, {"jsonrpc": "2.0", "method": "initialize", "id": 1})
    assert resp.status == 200
    assert body["result"]["serverInfo"]["name"] == "ggrmcp-amd"
    assert body["id"] == 1


def test_tools_list(env):
    _, chained, _ = env
    resp, body = post(chained, {"jsonrpc": "2.0", "method": "tools/list", "id": 2})
    tools = {t["name"]: t for t in body["result"]["tools"]}
    assert "hello_helloservice_sayhello" in tools
    t = tools["hello_helloservice_sayhello"]
    assert t["inputSchema"]["properties"]["name"] == {"type": "string"}
    assert t["description"] == "SayHello returns a greeting for the given name."


def test_tools_call_roundtrip(env):
    _, chained, _ = env
    resp, body = post(
        chained,
        {
            "jsonrpc": "2.0",
            "method": "tools/call",
            "id": 3,
            "params": {
                "name": "hello_helloservice_sayhello",
                "arguments": {"name": "world"},
            },
        },
    )
    assert resp.status == 200
    result = body["result"]
    assert result["isError"] is False
    assert json.loads(result["content"][0]["text"]) == {"message": "Hello, world!"}


def test_tools_call_grpc_error_is_tool_error(env):
    _, chained, _ = env
    _, body = post(
        chained,
        {
            "jsonrpc": "2.0",
            "method": "tools/call",
            "id": 4,
            "params": {"name": "hello_helloservice_sayhello", "arguments": {"name": "error"}},
        },
    )
    result = body["result"]
    assert result["isError"] is True
    assert "INVALID_ARGUMENT" in result["content"][0]["text"]


def test_tools_call_unknown_tool(env):
    _, chained, _ = env
    _, body = post(
        chained,
        {"jsonrpc": "2.0", "method": "tools/call", "id": 5, "params": {"name": "missing_tool"}},
    )
    assert body["error"]["code"] == -32601


def test_parse_error(env):
    _, chained, _ = env
    req = Request(
        method="POST", path="/", headers={"content-type": "application/json"}, body=b"{nope"
    )
    resp = asyncio.run(chained(req))
    body = json.loads(resp.body)
    assert resp.status == 200  # JSON-RPC errors are HTTP 200 (handler.go:311)
    assert body["error"]["code"] == -32700


def test_method_not_found(env):
    _, chained, _ = env
    _, body = post(chained, {"jsonrpc": "2.0", "method": "bogus/method", "id": 6})
    assert body["error"]["code"] == -32601


def test_invalid_request(env):
    _, chained, _ = env
    _, body = post(chained, {"jsonrpc": "1.0", "method": "tools/list", "id": 7})
    assert body["error"]["code"] == -32600


def test_notification_accepted(env):
    _, chained, _ = env
    resp, _ = post(chained, {"jsonrpc": "2.0", "method": "notifications/initialized"})
    assert resp.status == 202


def test_session_reuse(env):
    _, chained, _ = env
    resp1, _ = post(chained, {"jsonrpc": "2.0", "method": "initialize", "id": 1})
    sid = resp1.headers["Mcp-Session-Id"]
    resp2, _ = post(
        chained,
        {"jsonrpc": "2.0", "method": "initialize", "id": 2},
        headers={"mcp-session-id": sid},
    )
    assert resp2.headers["Mcp-Session-Id"] == sid


def test_header_forwarding_filters(env):
    handler, chained, _ = env
    captured = {}
    real_invoke = handler.invoker.invoke

    async def spy(tool_name, args_json, headers, timeout_s):
        captured["headers"] = headers
        return await real_invoke(tool_name, args_json, headers, timeout_s)

    handler.invoker.invoke = spy
    try:
        post(
            chained,
            {
                "jsonrpc": "2.0",
                "method": "tools/call",
                "id": 9,
                "params": {"name": "hello_helloservice_sayhello", "arguments": {"name": "h"}},
            },
            headers={
                "authorization": "Bearer tok",
                "cookie": "evil=1",
                "x-trace-id": "t-1",
                "x-custom": "nope",
            },
        )
    finally:
        handler.invoker.invoke = real_invoke
    fwd = captured["headers"]
    assert fwd.get("authorization") == "Bearer tok"
    assert fwd.get("x-trace-id") == "t-1"
    assert "cookie" not in fwd
    assert "x-custom" not in fwd
    assert "mcp-session-id" not in fwd


def test_streaming_tool_call_returns_chunks(env):
    _, chained, _ = env
    _, body = post(
        chained,
        {
            "jsonrpc": "2.0",
            "method": "tools/call",
            "id": 10,
            "params": {
                "name": "complex_nodeservice_streamnodes",
                "arguments": {"root": {"value": "x"}, "depth": 3},
            },
        },
    )
    result = body["result"]
    assert result["isError"] is False
    assert len(result["content"]) == 3


def test_prompts_and_resources_empty(env):
    _, chained, _ = env
    _, body = post(chained, {"jsonrpc": "2.0", "method": "prompts/list", "id": 11})
    assert body["result"] == {"prompts": []}
    _, body = post(chained, {"jsonrpc": "2.0", "method": "resources/list", "id": 12})
    assert body["result"] == {"resources": []}


def test_health(env):
    _, chained, _ = env
    resp = asyncio.run(chained(Request(method="GET", path="/health")))
    assert resp.status == 200
    body = json.loads(resp.body)
    assert body["status"] == "healthy"
    assert body["methodCount"] >= 5


def test_metrics(env):
    _, chained, recorder = env
    resp = asyncio.run(chained(Request(method="GET", path="/metrics")))
    body = json.loads(resp.body)
    assert body["methodCount"] >= 5
    assert "sessions" in body
    assert recorder.snapshot()["httpRequests"] > 0


def test_content_type_rejected(env):
    _, chained, _ = env
    req = Request(
        method="POST", path="/", headers={"content-type": "text/xml"}, body=b"<x/>"
    )
    resp = asyncio.run(chained(req))
    assert resp.status == 415


def test_body_cap(env):
    _, chained, _ = env
    req = Request(
        method="POST",
        path="/",
        headers={"content-type": "application/json"},
        body=b"x" * (1024 * 1024 + 1),
    )
    resp = asyncio.run(chained(req))
    assert resp.status == 413


def test_blocked_session_rejected(env):
    handler, chained, _ = env
    resp1, _ = post(chained, {"jsonrpc": "2.0", "method": "initialize", "id": 1})
    sid = resp1.headers["Mcp-Session-Id"]
    handler.sessions.block(sid)
    try:
        _, body = post(
            chained,
            {
                "jsonrpc": "2.0",
                "method": "tools/call",
                "id": 2,
                "params": {"name": "hello_helloservice_sayhello", "arguments": {"name": "x"}},
            },
            headers={"mcp-session-id": sid},
        )
        assert body["error"]["code"] == -32600
        assert "blocked" in body["error"]["message"]
    finally:
        handler.sessions.unblock(sid)
