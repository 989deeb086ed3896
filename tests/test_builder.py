"""Tool/schema builder tests — mirror reference pkg/tools/builder_test.go
coverage: recursive types, oneof, enum, map, WKT, circular $ref."""

import pytest

from examples.protos import ALL_FDPS
from ggrmcp_amd.descriptors.loader import build_pool, extract_method_infos
from ggrmcp_amd.tools import MCPToolBuilder, build_comment_index


@pytest.fixture(scope="module")
def env():
    pool = build_pool(ALL_FDPS)
    infos = {m.tool_name(): m for m in extract_method_infos(ALL_FDPS, pool, compat_names=False)}
    builder = MCPToolBuilder(build_comment_index(ALL_FDPS))
    return pool, infos, builder


def test_build_simple_tool(env):
    _, infos, builder = env
    tool = builder.build_tool(infos["hello_helloservice_sayhello"])
    assert tool.name == "hello_helloservice_sayhello"
    assert tool.description == "SayHello returns a greeting for the given name."
    schema = tool.input_schema
    assert schema["type"] == "object"
    assert schema["properties"]["name"] == {"type": "string"}
    assert schema["required"] == ["name"]
    assert tool.output_schema["properties"]["message"] == {"type": "string"}


def test_description_fallback(env):
    _, infos, builder = env
    tool = builder.build_tool(infos["complex_userservice_getuser"])
    assert tool.description == (
        "Calls the GetUser method of the complex.UserService service"
    )


def test_enum_and_wkt_and_int64(env):
    _, infos, builder = env
    tool = builder.build_tool(infos["complex_userservice_getuser"])
    out = tool.output_schema
    status = out["properties"]["status"]
    assert status["type"] == "string"
    assert status["enum"] == ["STATUS_UNSPECIFIED", "STATUS_ACTIVE", "STATUS_INACTIVE"]
    assert out["properties"]["createdAt"] == {"type": "string", "format": "date-time"}
    assert out["properties"]["score"] == {"type": "integer", "format": "int64"}
    assert out["properties"]["tags"] == {"type": "array", "items": {"type": "string"}}
    assert out["properties"]["avatar"] == {"type": "string", "format": "byte"}
    # presence: scalar proto3 fields are required, message fields are not
    assert "userId" in out["required"]
    assert "createdAt" not in out.get("required", [])


def test_oneof_and_map(env):
    _, infos, builder = env
    tool = builder.build_tool(infos["complex_documentservice_putdocument"])
    schema = tool.input_schema
    # oneof: text/binary mutually exclusive (builder.go:214-253)
    assert "allOf" in schema
    oneof = schema["allOf"][0]["oneOf"]
    assert {"required": ["text"]} in oneof
    assert {"required": ["binary"]} in oneof
    # map -> patternProperties (builder.go:284-297)
    meta = schema["properties"]["metadata"]
    assert meta["patternProperties"] == {".*": {"type": "string"}}
    # oneof members are not in required
    assert "text" not in schema.get("required", [])


def test_recursive_message_uses_ref(env):
    _, infos, builder = env
    tool = builder.build_tool(infos["complex_nodeservice_echo"])
    schema = tool.input_schema
    root = schema["properties"]["root"]
    children = root["properties"]["children"]
    assert children["items"] == {"$ref": "#/definitions/complex.Node"}
    defs = schema["definitions"]
    assert "complex.Node" in defs
    assert defs["complex.Node"]["properties"]["value"] == {"type": "string"}


def test_build_tools_skips_client_streaming_keeps_server_streaming(env):
    _, infos, builder = env
    from ggrmcp_amd.types import MethodInfo

    tools = builder.build_tools(infos.values())
    names = {t.name for t in tools}
    assert "complex_nodeservice_streamnodes" in names  # extension over reference
    cs = MethodInfo(
        service_name="x.Y",
        method_name="Up",
        input_descriptor=infos["hello_helloservice_sayhello"].input_descriptor,
        output_descriptor=infos["hello_helloservice_sayhello"].output_descriptor,
        is_client_streaming=True,
    )
    assert builder.build_tools([cs]) == []


def test_schema_cache_works(env):
    _, infos, builder = env
    builder.set_cache_key(1)
    s1 = builder.extract_message_schema(infos["hello_helloservice_sayhello"].input_descriptor)
    s2 = builder.extract_message_schema(infos["hello_helloservice_sayhello"].input_descriptor)
    assert s1 is s2  # cached, not rebuilt (improvement over builder.go:18,29)
    builder.set_cache_key(2)
    s3 = builder.extract_message_schema(infos["hello_helloservice_sayhello"].input_descriptor)
    assert s3 is not s1


def test_comment_index(env):
    pool, _, _ = env
    idx = build_comment_index(ALL_FDPS)
    assert idx["hello.HelloService.SayHello"].startswith("SayHello returns")
    assert idx["hello.HelloRequest"].startswith("The request")


def test_circular_proto_dependency_rejected():
    # loader.go:67-134's recursive processFile would loop; ours raises
    # (loader.py build_pool cycle detection)
    from google.protobuf import descriptor_pb2

    a = descriptor_pb2.FileDescriptorProto(name="a.proto", package="a",
                                           dependency=["b.proto"])
    b = descriptor_pb2.FileDescriptorProto(name="b.proto", package="b",
                                           dependency=["a.proto"])
    with pytest.raises(ValueError, match="circular"):
        build_pool([a, b])
