"""Full GPU pipeline tests (gfx950): process_batch end-to-end over the
native transport + native backend, including error slots and fallbacks."""

import json
import random

import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def pipeline_env():
    from google.protobuf import descriptor_pb2

    from examples.protos import ALL_FDPS
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer
    from ggrmcp_amd.backend.native_invoker import NativeWireClient, load_module
    from ggrmcp_amd.config import Config
    from ggrmcp_amd.engine.batch import GpuPipeline
    from ggrmcp_amd.utils.synthetic import synthetic_fdp

    mod = load_module()
    srv = mod.Server("127.0.0.1:0")
    srv.add_route("/hello.HelloService/SayHello", "hello")
    srv.add_route("/bench.EchoService/Echo", "echo")
    srv.add_route("/bench.EchoService/StreamEcho", "stream_echo")
    bound = srv.start()

    cfg = Config.default()
    host, _, port = bound.rpartition(":")
    cfg.grpc.host, cfg.grpc.port = host, int(port)
    d = ServiceDiscoverer(cfg)
    fdset = descriptor_pb2.FileDescriptorSet()
    fdset.file.extend(ALL_FDPS + [synthetic_fdp()])
    d.load_descriptor_blob(fdset.SerializeToString())
    d.connections[0].connect(timeout_s=15)
    wire = NativeWireClient(bound, connections=2)
    pipeline = GpuPipeline(d, cfg, device=0, wire_clients=[wire])
    yield pipeline
    wire.close()
    d.close()
    srv.stop()


def _body(tool, args, rid):
    return json.dumps(
        {"jsonrpc": "2.0", "id": rid, "method": "tools/call",
         "params": {"name": tool, "arguments": args}}
    ).encode()


def test_process_batch_roundtrip(pipeline_env):
    pipeline = pipeline_env
    bodies = [
        _body("hello_helloservice_sayhello", {"name": f"u{i}"}, i) for i in range(32)
    ]
    out = pipeline.process_batch(bodies, timeout_s=15.0)
    assert len(out) == 32
    for i, raw in enumerate(out):
        resp = json.loads(raw)
        assert resp["id"] == i
        assert resp["result"]["isError"] is False
        inner = json.loads(resp["result"]["content"][0]["text"])
        assert inner == {"message": f"Hello, u{i}!"}
    assert pipeline.engine.stats.gpu_ok >= 32
    # per-stage split (SURVEY §5 /metrics deliverable): the native span
    # executor must attribute encode/decode time, not book it all as invoke
    snap = pipeline.engine.stats.snapshot()
    assert snap["encodeMs"] > 0
    assert snap["decodeMs"] > 0
    assert snap["invokeMs"] > 0


def test_process_batch_mixed_errors(pipeline_env):
    pipeline = pipeline_env
    bodies = [
        _body("hello_helloservice_sayhello", {"name": "ok"}, 1),
        _body("hello_helloservice_sayhello", {"name": "error"}, 2),  # gRPC error
        _body("missing_tool", {}, 3),                                # -32601
        b"{broken json",                                             # -32700
        json.dumps({"jsonrpc": "2.0", "id": 5, "method": "tools/list"}).encode(),
        _body("hello_helloservice_sayhello", {"nope": 1}, 6),        # -32602
    ]
    out = pipeline.process_batch(bodies, timeout_s=15.0)
    r = [json.loads(x) for x in out]
    assert r[0]["result"]["isError"] is False
    assert r[1]["result"]["isError"] is True
    assert "INVALID_ARGUMENT" in r[1]["result"]["content"][0]["text"]
    assert r[2]["error"]["code"] == -32601
    assert r[3]["error"]["code"] == -32700
    assert r[4]["error"]["code"] == -32601  # tools/list not on batch path
    assert r[5]["error"]["code"] == -32602


def test_process_batch_wide64(pipeline_env):
    from ggrmcp_amd.utils.synthetic import wide_payload

    pipeline = pipeline_env
    rng = random.Random(3)
    bodies = [
        _body("bench_echoservice_echo", wide_payload(rng), i) for i in range(16)
    ]
    out = pipeline.process_batch(bodies, timeout_s=15.0)
    for i, raw in enumerate(out):
        resp = json.loads(raw)
        assert resp["result"]["isError"] is False, resp
        inner = json.loads(resp["result"]["content"][0]["text"])
        assert inner["nested"]["value"] == "42"


def test_process_batch_ids_preserved(pipeline_env):
    pipeline = pipeline_env
    bodies = [
        _body("hello_helloservice_sayhello", {"name": "a"}, "string-id"),
        _body("hello_helloservice_sayhello", {"name": "b"}, 3.5),
        json.dumps({"jsonrpc": "2.0", "id": None, "method": "tools/call",
                    "params": {"name": "hello_helloservice_sayhello",
                               "arguments": {"name": "c"}}}).encode(),
    ]
    out = pipeline.process_batch(bodies, timeout_s=15.0)
    assert json.loads(out[0])["id"] == "string-id"
    assert json.loads(out[1])["id"] == 3.5
    assert json.loads(out[2])["id"] is None


@pytest.fixture(scope="module")
def grpcio_pipeline_env():
    """Pipeline over the python grpcio backend (has server-streaming)."""
    from google.protobuf import descriptor_pb2

    from examples.bench_backend import serve
    from examples.protos import ALL_FDPS
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer
    from ggrmcp_amd.config import Config
    from ggrmcp_amd.engine.batch import GpuPipeline
    from ggrmcp_amd.utils.synthetic import synthetic_fdp

    srv, bound = serve("127.0.0.1:0")
    cfg = Config.default()
    host, _, port = bound.rpartition(":")
    cfg.grpc.host, cfg.grpc.port = host, int(port)
    d = ServiceDiscoverer(cfg)
    fdset = descriptor_pb2.FileDescriptorSet()
    fdset.file.extend(ALL_FDPS + [synthetic_fdp()])
    d.load_descriptor_blob(fdset.SerializeToString())
    d.connections[0].connect(timeout_s=15)
    pipeline = GpuPipeline(d, cfg, device=0)
    yield pipeline
    d.close()
    srv.stop(grace=None)


def test_process_batch_streaming_gpu_decode(grpcio_pipeline_env):
    """Server-streaming slots decode their chunks in one GPU batch
    (the reference rejects streaming outright, discovery.go:354-356)."""
    pipeline = grpcio_pipeline_env
    bodies = [
        _body("complex_nodeservice_streamnodes", {"root": {"value": "s"}, "depth": 5}, 1),
        _body("hello_helloservice_sayhello", {"name": "u"}, 2),
        _body("complex_nodeservice_streamnodes", {"root": {"value": "t"}, "depth": 3}, 3),
    ]
    before_ok = pipeline.engine.stats.gpu_ok
    out = pipeline.process_batch(bodies, timeout_s=15.0)
    r0, r1, r2 = (json.loads(x) for x in out)
    assert r0["result"]["isError"] is False
    assert len(r0["result"]["content"]) == 5
    chunks = [json.loads(c["text"]) for c in r0["result"]["content"]]
    assert [c.get("depth", 0) for c in chunks] == [0, 1, 2, 3, 4]
    assert all(c["root"]["value"] == "s" for c in chunks)
    assert r1["result"]["isError"] is False
    assert len(r2["result"]["content"]) == 3
    # streams took the GPU decode path, not the host fallback
    assert pipeline.engine.stats.gpu_ok >= before_ok + 3
    assert pipeline.engine.stats.host_fallbacks == 0


def test_native_mixed_unary_and_stream(pipeline_env):
    """Unary + server-streaming slots in ONE batch through the native span:
    the stream-chunk decode must not clobber the unary responses
    (regression: both shared the engine's pinned output buffer)."""
    pipeline = pipeline_env
    bodies = []
    for i in range(12):
        if i % 3 == 2:
            bodies.append(_body("bench_echoservice_streamecho",
                                {"f01String": f"s{i}", "f02Int32": 5}, i))
        else:
            bodies.append(_body("hello_helloservice_sayhello",
                                {"name": f"u{i}"}, i))
    out = pipeline.process_batch(bodies, timeout_s=15.0)
    for i, raw in enumerate(out):
        resp = json.loads(raw)
        assert resp["id"] == i, resp
        assert resp["result"]["isError"] is False, resp
        if i % 3 == 2:
            assert len(resp["result"]["content"]) == 5
            inner = json.loads(resp["result"]["content"][0]["text"])
            assert inner["f01String"] == f"s{i}"
        else:
            inner = json.loads(resp["result"]["content"][0]["text"])
            assert inner == {"message": f"Hello, u{i}!"}


def test_badutf8_response_single_invoke():
    """End-to-end decode-stage fallback: backend returns a HelloResponse with
    invalid UTF-8; the GPU decode rejects it (strict UTF-8), and the gateway
    transcodes the received bytes on the host WITHOUT re-invoking the RPC
    (server request_count stays exact — VERDICT r1 item 2)."""
    from google.protobuf import descriptor_pb2

    from examples.protos import ALL_FDPS
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer
    from ggrmcp_amd.backend.native_invoker import NativeWireClient, load_module
    from ggrmcp_amd.config import Config
    from ggrmcp_amd.engine.batch import GpuPipeline

    mod = load_module()
    srv = mod.Server("127.0.0.1:0")
    srv.add_route("/hello.HelloService/SayHello", "hello")
    bound = srv.start()
    wire = None
    d = None
    try:
        cfg = Config.default()
        host, _, port = bound.rpartition(":")
        cfg.grpc.host, cfg.grpc.port = host, int(port)
        d = ServiceDiscoverer(cfg)
        fdset = descriptor_pb2.FileDescriptorSet()
        fdset.file.extend(ALL_FDPS)
        d.load_descriptor_blob(fdset.SerializeToString())
        d.connections[0].connect(timeout_s=15)
        wire = NativeWireClient(bound, connections=1)
        pipeline = GpuPipeline(d, cfg, device=0, wire_clients=[wire])

        bodies = [
            _body("hello_helloservice_sayhello", {"name": "ok"}, 1),
            _body("hello_helloservice_sayhello", {"name": "badutf8"}, 2),
        ]
        before = srv.request_count()
        out = pipeline.process_batch(bodies, timeout_s=15.0)
        after = srv.request_count()
        assert after - before == 2, "each request must be invoked exactly once"

        ok = json.loads(out[0])
        assert ok["id"] == 1 and ok["result"]["isError"] is False
        bad = json.loads(out[1])
        assert bad["id"] == 2
        # invalid UTF-8: both GPU and protojson oracle reject -> internal
        # error envelope, no isError=false result fabricated from garbage
        assert "error" in bad
        assert pipeline.engine.stats.host_fallbacks >= 1
    finally:
        if wire is not None:
            wire.close()
        if d is not None:
            d.close()
        srv.stop()
