"""Shared type tests — mirror reference pkg/types/service.go tool naming."""

from ggrmcp_amd.types import MethodInfo, generate_tool_name


def test_generate_tool_name():
    assert (
        generate_tool_name("hello.HelloService", "SayHello")
        == "hello_helloservice_sayhello"
    )
    assert (
        generate_tool_name("com.example.UserService", "GetUser")
        == "com_example_userservice_getuser"
    )


def test_method_info_paths():
    mi = MethodInfo(service_name="hello.HelloService", method_name="SayHello")
    assert mi.full_method_path == "/hello.HelloService/SayHello"
    assert mi.tool_name() == "hello_helloservice_sayhello"
    assert not mi.is_streaming
    mi.is_server_streaming = True
    assert mi.is_streaming


def test_mcp_wire_shapes():
    from ggrmcp_amd.mcp import (
        JSONRPCResponse,
        RPCError,
        TextContent,
        Tool,
        ToolCallResult,
        initialization_result,
    )

    r = JSONRPCResponse(id=1, result={"ok": True})
    assert r.to_dict() == {"jsonrpc": "2.0", "id": 1, "result": {"ok": True}}
    e = JSONRPCResponse(id="x", error=RPCError(-32601, "not found"))
    assert e.to_dict()["error"] == {"code": -32601, "message": "not found"}
    t = Tool(name="n", description="d", input_schema={"type": "object"})
    assert t.to_dict()["inputSchema"] == {"type": "object"}
    res = ToolCallResult(content=[TextContent("hi")], is_error=False)
    assert res.to_dict() == {
        "content": [{"type": "text", "text": "hi"}],
        "isError": False,
    }
    init = initialization_result()
    assert init["protocolVersion"] == "2024-11-05"
    assert set(init["capabilities"]) == {"tools", "prompts", "resources"}
