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


def test_descriptor_set_file_discovery(tmp_path):
    """--descriptor route: .binpb file -> loader -> tool map
    (discovery.go:101-119 descriptor-set-first path)."""
    from google.protobuf import descriptor_pb2

    from examples.protos import ALL_FDPS
    from ggrmcp_amd.config import Config
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer
    from ggrmcp_amd.utils.synthetic import synthetic_fdp

    fdset = descriptor_pb2.FileDescriptorSet()
    fdset.file.extend(ALL_FDPS + [synthetic_fdp()])
    path = tmp_path / "services.binpb"
    path.write_bytes(fdset.SerializeToString())

    cfg = Config.default()
    cfg.descriptor_set.enabled = True
    cfg.descriptor_set.path = str(path)
    d = ServiceDiscoverer(cfg)
    tools = d.discover()  # no connection needed for the descriptor path
    assert "hello_helloservice_sayhello" in tools
    assert "bench_echoservice_echo" in tools
    assert tools["bench_echoservice_streamecho"].is_server_streaming


def test_descriptor_set_missing_file_falls_back(tmp_path, backend):
    """Bad descriptor path logs a warning and falls back to reflection
    (discovery.go:107-111)."""
    from ggrmcp_amd.config import Config
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer

    target = backend
    cfg = Config.default()
    host, _, port = target.rpartition(":")
    cfg.grpc.host, cfg.grpc.port = host, int(port)
    cfg.descriptor_set.enabled = True
    cfg.descriptor_set.path = str(tmp_path / "nope.binpb")
    d = ServiceDiscoverer(cfg)
    d.connect(timeout_s=10)
    tools = d.discover()
    assert "hello_helloservice_sayhello" in tools  # via reflection
    d.close()


def test_reconnect_with_retry(backend):
    """discovery.go:186-235: bounded reconnect + full rediscovery."""
    from ggrmcp_amd.config import Config
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer

    target = backend
    cfg = Config.default()
    host, _, port = target.rpartition(":")
    cfg.grpc.host, cfg.grpc.port = host, int(port)
    d = ServiceDiscoverer(cfg)
    d.connect(timeout_s=10)
    d.discover()
    v1 = d.tools_version
    assert d.reconnect_with_retry(attempts=2, delay_s=0.1)
    assert d.tools_version > v1  # rediscovered + atomic republish
    assert "hello_helloservice_sayhello" in d.tools
    d.close()


def test_malformed_descriptor_blob_rejected(discoverer):
    """A hostile/corrupt FileDescriptorSet blob must raise a clean error
    (protobuf DecodeError surface), never publish a partial tool map."""
    import pytest as _pytest

    before = dict(discoverer.tools)
    with _pytest.raises(Exception):
        discoverer.load_descriptor_blob(b"\x00\xff garbage \x01\x02" * 20)
    assert discoverer.tools == before  # atomic publish: map unchanged


def test_corrupt_descriptor_set_file_falls_back(tmp_path, backend):
    """descriptor_set.path pointing at corrupt bytes behaves like the
    missing-file case: warn + reflection fallback (discovery.go:107-111)."""
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer
    from ggrmcp_amd.config import Config

    bad = tmp_path / "bad.binpb"
    bad.write_bytes(b"\xde\xad\xbe\xef" * 64)
    cfg = Config.default()
    host, _, port = backend.rpartition(":")
    cfg.grpc.host, cfg.grpc.port = host, int(port)
    cfg.descriptor_set.enabled = True
    cfg.descriptor_set.path = str(bad)
    d = ServiceDiscoverer(cfg)
    try:
        d.connect(timeout_s=15)
        d.discover()
        assert "hello_helloservice_sayhello" in d.tools  # via reflection
    finally:
        d.close()
