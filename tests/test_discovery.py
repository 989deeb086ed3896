"""Discovery + invocation integration tests against an in-process gRPC
backend — the analog of the reference's bufconn TestEnvironment
(tests/test_utils.go:41-114) using a loopback TCP port."""

import json

import pytest

from examples.hello_service import serve
from ggrmcp_amd.backend.discovery import (
    STREAMING_UNSUPPORTED_MSG,
    MethodNotFoundError,
    ServiceDiscoverer,
)
from ggrmcp_amd.config import Config


@pytest.fixture(scope="module")
def backend():
    server, target = serve("127.0.0.1:0")
    yield target
    server.stop(grace=None)


@pytest.fixture(scope="module")
def discoverer(backend):
    cfg = Config.default()
    host, _, port = backend.rpartition(":")
    cfg.grpc.host = host
    cfg.grpc.port = int(port)
    d = ServiceDiscoverer(cfg)
    d.connect(timeout_s=10)
    d.discover()
    yield d
    d.close()


def test_discovers_all_tools(discoverer):
    tools = discoverer.tools
    assert "hello_helloservice_sayhello" in tools
    assert "complex_userservice_getuser" in tools
    assert "complex_documentservice_putdocument" in tools
    assert "complex_nodeservice_echo" in tools
    assert "complex_nodeservice_streamnodes" in tools
    mi = tools["hello_helloservice_sayhello"]
    assert mi.input_descriptor.full_name == "hello.HelloRequest"
    assert mi.description == "SayHello returns a greeting for the given name."
    streaming = tools["complex_nodeservice_streamnodes"]
    assert streaming.is_server_streaming


def test_internal_services_filtered(discoverer):
    assert not any(
        t.startswith("grpc_reflection") for t in discoverer.tools
    )


def test_invoke_hello(discoverer):
    out = discoverer.invoke_method_by_tool(
        "hello_helloservice_sayhello", json.dumps({"name": "MI355X"})
    )
    assert json.loads(out) == {"message": "Hello, MI355X!"}


def test_invoke_complex_types(discoverer):
    out = discoverer.invoke_method_by_tool(
        "complex_userservice_getuser", json.dumps({"userId": "42"})
    )
    data = json.loads(out)
    assert data["userId"] == "42"
    assert data["status"] == "STATUS_ACTIVE"  # enum as name
    assert data["score"] == "9007199254740993"  # int64 as string
    assert data["createdAt"] == "2023-11-14T22:13:20.123Z"  # Timestamp WKT
    assert data["tags"] == ["alpha", "beta"]


def test_invoke_oneof_and_map(discoverer):
    doc = {"id": "d1", "text": "hello", "metadata": {"k": "v", "k2": "v2"}}
    out = json.loads(
        discoverer.invoke_method_by_tool(
            "complex_documentservice_putdocument", json.dumps(doc)
        )
    )
    assert out == doc


def test_invoke_recursive(discoverer):
    tree = {"root": {"value": "a", "children": [{"value": "b"}, {"value": "c"}]}}
    out = json.loads(
        discoverer.invoke_method_by_tool("complex_nodeservice_echo", json.dumps(tree))
    )
    assert out == tree


def test_invoke_error_mapping(discoverer):
    import grpc

    with pytest.raises(grpc.RpcError) as ei:
        discoverer.invoke_method_by_tool(
            "hello_helloservice_sayhello", json.dumps({"name": "error"})
        )
    assert ei.value.code() == grpc.StatusCode.INVALID_ARGUMENT


def test_streaming_rejected_on_unary_path(discoverer):
    with pytest.raises(ValueError, match=STREAMING_UNSUPPORTED_MSG):
        discoverer.invoke_method_by_tool(
            "complex_nodeservice_streamnodes", json.dumps({"depth": 3})
        )


def test_streaming_supported_via_invoke_streaming(discoverer):
    chunks = list(
        discoverer.invoke_streaming(
            "complex_nodeservice_streamnodes",
            json.dumps({"root": {"value": "x"}, "depth": 3}),
        )
    )
    assert len(chunks) == 3
    assert json.loads(chunks[1])["depth"] == 1


def test_unknown_tool(discoverer):
    with pytest.raises(MethodNotFoundError):
        discoverer.invoke_method_by_tool("nope", "{}")


def test_health_and_stats(discoverer):
    assert discoverer.health_check()
    stats = discoverer.stats()
    assert stats["methodCount"] >= 5
    assert stats["serviceCount"] == 4
    assert stats["isConnected"]


def test_descriptor_blob_roundtrip(discoverer, backend):
    blob = discoverer.descriptor_blob()
    assert blob
    cfg = Config.default()
    host, _, port = backend.rpartition(":")
    cfg.grpc.host, cfg.grpc.port = host, int(port)
    other = ServiceDiscoverer(cfg)
    other.load_descriptor_blob(blob)
    assert set(other.tools) == set(discoverer.tools)
