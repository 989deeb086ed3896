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
    yield gw, port_http, pipeline, srv
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


def _gpu_ok(gw, pipeline):
    """GPU-completed requests across both serving paths: the Python
    batch_cb pipeline and the GIL-free native span (C++ counters)."""
    n = pipeline.engine.stats.gpu_ok
    if getattr(gw, "_span_engines", None):
        n += gw._fe.native_stats()["gpuOk"]
    return n


def test_serving_roundtrip_on_gpu(native_gateway):
    gw, port, pipeline = native_gateway[:3]
    before = _gpu_ok(gw, pipeline)
    body = json.dumps({"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                       "params": {"name": "hello_helloservice_sayhello",
                                  "arguments": {"name": "serve"}}})
    status, resp, sid = _post(port, body)
    assert status == 200 and sid
    inner = json.loads(resp["result"]["content"][0]["text"])
    assert inner == {"message": "Hello, serve!"}
    assert _gpu_ok(gw, pipeline) > before  # GPU path, not fallback


def test_native_span_active_on_gpu(native_gateway):
    """The serving hot path must be the GIL-free C++ span (span_api.h) —
    not the Python batch_cb fallback (make the serving number
    the native number)."""
    gw, port, pipeline = native_gateway[:3]
    assert gw._span_engines, "native span must be enabled on the GPU path"
    st0 = gw._fe.native_stats()
    body = json.dumps({"jsonrpc": "2.0", "id": 11, "method": "tools/call",
                       "params": {"name": "hello_helloservice_sayhello",
                                  "arguments": {"name": "span"}}})
    status, resp, _ = _post(port, body)
    assert status == 200
    st1 = gw._fe.native_stats()
    assert st1["gpuOk"] > st0["gpuOk"]
    assert st1["requests"] > st0["requests"]
    assert st1["encodeMs"] > 0 and st1["decodeMs"] > 0


def test_native_span_grpc_error_envelope(native_gateway):
    """gRPC failure -> C++-assembled isError result (handler.go:252-259
    semantics) with the request id preserved from the kernel's id slot."""
    gw, port, pipeline = native_gateway[:3]
    body = json.dumps({"jsonrpc": "2.0", "id": "err-1", "method": "tools/call",
                       "params": {"name": "hello_helloservice_sayhello",
                                  "arguments": {"name": "error"}}})
    status, resp, _ = _post(port, body)
    assert status == 200
    assert resp["id"] == "err-1"
    assert resp["result"]["isError"] is True
    text = resp["result"]["content"][0]["text"]
    assert text.startswith("gRPC error INVALID_ARGUMENT")


def test_native_span_decode_fallback_single_invoke(native_gateway):
    """badutf8 response through the SERVING path: decode rejects, the
    fallback transcodes the delivered wire — zero duplicate invokes."""
    gw, port, pipeline = native_gateway[:3]
    body = json.dumps({"jsonrpc": "2.0", "id": 21, "method": "tools/call",
                       "params": {"name": "hello_helloservice_sayhello",
                                  "arguments": {"name": "badutf8"}}})
    srv = native_gateway[3]
    before = srv.request_count()
    status, resp, _ = _post(port, body)
    after = srv.request_count()
    assert status == 200
    assert resp["id"] == 21
    # both GPU and the protojson oracle reject invalid UTF-8 -> error
    # envelope, produced WITHOUT a second invoke
    assert "error" in resp
    assert after - before == 1, "fallback must not re-invoke" 


def test_serving_concurrent_sessions(native_gateway):
    gw, port, pipeline = native_gateway[:3]
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
    gw, port, pipeline = native_gateway[:3]
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

    gw, port, pipeline = native_gateway[:3]
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


def test_serving_large_payloads_wg_kernels(native_gateway):
    """64 KB tool calls through the SERVING path: the workgroup-cooperative
    encode/decode kernels run inside the GIL-free span executor and the
    responses stay protojson-exact."""
    import random

    from ggrmcp_amd.engine.cpu_ref import CpuTranscoder
    from ggrmcp_amd.utils.synthetic import wide_payload

    gw, port, pipeline = native_gateway[:3]
    rng = random.Random(55)
    cpu = CpuTranscoder()
    d = gw.discoverer
    mi = d.tools["bench_echoservice_echo"]
    for i in range(4):
        args = wide_payload(rng, target_bytes=64 * 1024)
        body = json.dumps({"jsonrpc": "2.0", "id": 100 + i,
                           "method": "tools/call",
                           "params": {"name": "bench_echoservice_echo",
                                      "arguments": args}})
        status, resp, _ = _post(port, body)
        assert status == 200
        assert resp["id"] == 100 + i
        assert resp["result"]["isError"] is False, str(resp)[:300]
        inner = json.loads(resp["result"]["content"][0]["text"])
        wire = cpu.json_to_pb(mi.input_descriptor, json.dumps(args))
        oracle = json.loads(cpu.pb_to_json(mi.output_descriptor, wire))
        assert inner == oracle
