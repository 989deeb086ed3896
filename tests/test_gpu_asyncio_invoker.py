"""GPU coverage for the asyncio serving surface: the full-middleware
HTTPServer + BatchEngineInvoker (value-mode GPU transcode batches with the
window-collected async worker) — the path `python -m ggrmcp_amd` uses on a
GPU host.  The native frontend covers production serving; this pins the
parity surface's GPU wiring."""

import asyncio
import json
import threading

import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def asyncio_gateway():
    from google.protobuf import descriptor_pb2

    from examples.protos import ALL_FDPS
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer
    from ggrmcp_amd.backend.native_invoker import load_module
    from ggrmcp_amd.config import Config
    from ggrmcp_amd.engine.batch import BatchEngineInvoker
    from ggrmcp_amd.server.handler import MCPHandler
    from ggrmcp_amd.server.http import HTTPServer
    from ggrmcp_amd.server.middleware import default_middleware
    from ggrmcp_amd.utils.synthetic import synthetic_fdp

    mod = load_module()
    srv = mod.Server("127.0.0.1:0")
    srv.add_route("/hello.HelloService/SayHello", "hello")
    srv.add_route("/bench.EchoService/Echo", "echo")
    bound = srv.start()

    cfg = Config.default()
    host, _, port = bound.rpartition(":")
    cfg.grpc.host, cfg.grpc.port = host, int(port)
    cfg.server.rate_limit_rps = 100000
    cfg.server.rate_limit_burst = 100000
    d = ServiceDiscoverer(cfg)
    fdset = descriptor_pb2.FileDescriptorSet()
    fdset.file.extend(ALL_FDPS + [synthetic_fdp()])
    d.load_descriptor_blob(fdset.SerializeToString())
    d.connections[0].connect(timeout_s=15)

    from ggrmcp_amd.server.middleware import MetricsRecorder

    invoker = BatchEngineInvoker(d, cfg)
    handler = MCPHandler(d, config=cfg, invoker=invoker)
    http = HTTPServer(handler.handle,
                      middlewares=default_middleware(cfg.server,
                                                     MetricsRecorder()),
                      host="127.0.0.1", port=0)

    loop = asyncio.new_event_loop()
    started = threading.Event()

    def run():
        asyncio.set_event_loop(loop)
        loop.run_until_complete(http.start())
        started.set()
        loop.run_forever()

    t = threading.Thread(target=run, daemon=True)
    t.start()
    started.wait(30)
    yield http.port, invoker
    loop.call_soon_threadsafe(loop.stop)
    t.join(timeout=10)
    d.close()
    srv.stop()


def _post(port, body):
    import http.client

    conn = http.client.HTTPConnection("127.0.0.1", port, timeout=20)
    conn.request("POST", "/", body=body,
                 headers={"Content-Type": "application/json"})
    r = conn.getresponse()
    data = json.loads(r.read())
    conn.close()
    return r.status, data


def test_asyncio_gpu_tool_call(asyncio_gateway):
    port, invoker = asyncio_gateway
    before = invoker.pipeline.engine.stats.gpu_ok
    body = json.dumps({"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                       "params": {"name": "hello_helloservice_sayhello",
                                  "arguments": {"name": "aio"}}})
    status, resp = _post(port, body)
    assert status == 200
    inner = json.loads(resp["result"]["content"][0]["text"])
    assert inner == {"message": "Hello, aio!"}
    assert invoker.pipeline.engine.stats.gpu_ok > before, (
        "the asyncio invoker must take the GPU value-mode path")


def test_asyncio_gpu_concurrent_batching(asyncio_gateway):
    """Concurrent requests coalesce into value-mode GPU batches."""
    port, invoker = asyncio_gateway
    errs = []

    def worker(i):
        try:
            body = json.dumps(
                {"jsonrpc": "2.0", "id": i, "method": "tools/call",
                 "params": {"name": "hello_helloservice_sayhello",
                            "arguments": {"name": f"w{i}"}}})
            status, resp = _post(port, body)
            assert status == 200
            inner = json.loads(resp["result"]["content"][0]["text"])
            assert inner == {"message": f"Hello, w{i}!"}
        except Exception as e:  # pragma: no cover
            errs.append(e)

    ts = [threading.Thread(target=worker, args=(i,)) for i in range(32)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert not errs, errs
