"""End-to-end native serving on GPU: C++ HTTP reactor -> GpuPipeline
(k_json2pb / k_pb2json) -> native h2 transport -> native backend."""

import http.client
import json
import threading

import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def native_gateway():
    from google.protobuf import descriptor_pb2

    from examples.protos import ALL_FDPS
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer
    from ggrmcp_amd.backend.native_invoker import NativeWireClient, load_module
    from ggrmcp_amd.config import Config
    from ggrmcp_amd.engine.batch import GpuPipeline
    from ggrmcp_amd.server.native_http import NativeHTTPGateway
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
    cfg.server.rate_limit_rps = 1_000_000
    cfg.server.rate_limit_burst = 1_000_000
    d = ServiceDiscoverer(cfg)
    fdset = descriptor_pb2.FileDescriptorSet()
    fdset.file.extend(ALL_FDPS + [synthetic_fdp()])
    d.load_descriptor_blob(fdset.SerializeToString())
    d.connections[0].connect(timeout_s=15)
    wire = NativeWireClient(bound, connections=2)
    pipeline = GpuPipeline(d, cfg, device=0, wire_clients=[wire])
    gw = NativeHTTPGateway(pipeline, d, cfg)
    port_http = gw.start()
    yield gw, port_http, pipeline
    gw.stop()
    wire.close()
    d.close()
    srv.stop()


def _post(port, body, session=None):
    conn = http.client.HTTPConnection("127.0.0.1", port, timeout=15)
    headers = {"Content-Type": "application/json"}
    if session:
        headers["Mcp-Session-Id"] = session
    conn.request("POST", "/", body=body, headers=headers)
    r = conn.getresponse()
    data = r.read()
    sid = r.getheader("Mcp-Session-Id")
    conn.close()
    return r.status, json.loads(data), sid


def test_serving_roundtrip_on_gpu(native_gateway):
    gw, port, pipeline = native_gateway
    before = pipeline.engine.stats.gpu_ok
    body = json.dumps({"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                       "params": {"name": "hello_helloservice_sayhello",
                                  "arguments": {"name": "serve"}}})
    status, resp, sid = _post(port, body)
    assert status == 200 and sid
    inner = json.loads(resp["result"]["content"][0]["text"])
    assert inner == {"message": "Hello, serve!"}
    assert pipeline.engine.stats.gpu_ok > before  # GPU path, not fallback


def test_serving_concurrent_sessions(native_gateway):
    gw, port, pipeline = native_gateway
    errs = []

    def worker(t):
        try:
            conn = http.client.HTTPConnection("127.0.0.1", port, timeout=15)
            for i in range(10):
                body = json.dumps({"jsonrpc": "2.0", "id": f"{t}-{i}",
                                   "method": "tools/call",
                                   "params": {"name": "hello_helloservice_sayhello",
                                              "arguments": {"name": f"u{t}x{i}"}}})
                conn.request("POST", "/", body=body,
                             headers={"Content-Type": "application/json"})
                r = conn.getresponse()
                resp = json.loads(r.read())
                assert resp["id"] == f"{t}-{i}"
                inner = json.loads(resp["result"]["content"][0]["text"])
                assert inner == {"message": f"Hello, u{t}x{i}!"}
            conn.close()
        except Exception as e:  # pragma: no cover
            errs.append(e)

    ts = [threading.Thread(target=worker, args=(t,)) for t in range(24)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert not errs, errs


def test_serving_tools_list_and_errors(native_gateway):
    gw, port, pipeline = native_gateway
    status, resp, _ = _post(port, json.dumps(
        {"jsonrpc": "2.0", "id": 5, "method": "tools/list"}))
    assert status == 200
    names = [t["name"] for t in resp["result"]["tools"]]
    assert "hello_helloservice_sayhello" in names
    status, resp, _ = _post(port, json.dumps(
        {"jsonrpc": "2.0", "id": 6, "method": "tools/call",
         "params": {"name": "no_such_tool", "arguments": {}}}))
    assert resp["error"]["code"] == -32601
    status, resp, _ = _post(port, b"{broken")
    assert resp["error"]["code"] == -32700


def test_serving_soak_mixed(native_gateway):
    """Soak: concurrent sessions mixing unary, streaming, errors and
    malformed bodies through the reactor + native span; every response must
    match its request id and shape (order/captivity races would mispair)."""
    import random

    gw, port, pipeline = native_gateway
    errs = []

    def worker(t):
        rng = random.Random(t)
        try:
            conn = http.client.HTTPConnection("127.0.0.1", port, timeout=20)
            for i in range(25):
                kind = rng.randrange(5)
                rid = f"{t}-{i}"
                if kind == 0:
                    body = json.dumps({"jsonrpc": "2.0", "id": rid,
                                       "method": "tools/call",
                                       "params": {"name": "bench_echoservice_streamecho",
                                                  "arguments": {"f01String": rid,
                                                                "f02Int32": 3}}})
                elif kind == 1:
                    body = json.dumps({"jsonrpc": "2.0", "id": rid,
                                       "method": "tools/call",
                                       "params": {"name": "missing_tool",
                                                  "arguments": {}}})
                elif kind == 2:
                    body = json.dumps({"jsonrpc": "2.0", "id": rid,
                                       "method": "tools/list"})
                else:
                    body = json.dumps({"jsonrpc": "2.0", "id": rid,
                                       "method": "tools/call",
                                       "params": {"name": "hello_helloservice_sayhello",
                                                  "arguments": {"name": rid}}})
                conn.request("POST", "/", body=body,
                             headers={"Content-Type": "application/json"})
                r = conn.getresponse()
                resp = json.loads(r.read())
                assert resp["id"] == rid, (rid, resp)
                if kind == 0:
                    assert len(resp["result"]["content"]) == 3
                    inner = json.loads(resp["result"]["content"][1]["text"])
                    assert inner["f01String"] == rid
                elif kind == 1:
                    assert resp["error"]["code"] == -32601
                elif kind == 2:
                    assert "tools" in resp["result"]
                else:
                    inner = json.loads(resp["result"]["content"][0]["text"])
                    assert inner == {"message": f"Hello, {rid}!"}
            conn.close()
        except Exception as e:  # pragma: no cover
            errs.append((t, repr(e)))

    ts = [threading.Thread(target=worker, args=(t,)) for t in range(32)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert not errs, errs[:3]
