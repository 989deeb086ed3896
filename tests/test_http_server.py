"""End-to-end test over real TCP: demo backend -> gateway (full wiring via
cli.build_gateway) -> asyncio HTTP server -> httpx client. The analog of the
reference CI's live smoke (ci.yml:146-209) + httptest integration tests."""

import asyncio
import json
import threading

import httpx
import pytest

from examples.hello_service import serve
from ggrmcp_amd.cli import build_config, parse_args
from ggrmcp_amd.server.http import HTTPServer
from ggrmcp_amd.server.middleware import MetricsRecorder, default_middleware


@pytest.fixture(scope="module")
def gateway():
    backend, target = serve("127.0.0.1:0")
    host, _, port = target.rpartition(":")
    args = parse_args(
        ["--grpc-host", host, "--grpc-port", port, "--http-port", "0", "--no-gpu"]
    )
    cfg = build_config(args)
    cfg.server.rate_limit_rps = 1e6
    cfg.server.rate_limit_burst = 1000000

    from ggrmcp_amd.cli import build_gateway

    handler, discoverer = build_gateway(cfg)

    loop = asyncio.new_event_loop()
    server = HTTPServer(
        handler.handle,
        middlewares=default_middleware(cfg.server, MetricsRecorder()),
        host="127.0.0.1",
        port=0,
        max_body_bytes=cfg.server.max_body_bytes,
    )
    started = threading.Event()

    def run_loop():
        asyncio.set_event_loop(loop)

        async def main():
            await server.start()
            started.set()
            await asyncio.Event().wait()

        try:
            loop.run_until_complete(main())
        except RuntimeError:
            pass

    t = threading.Thread(target=run_loop, daemon=True)
    t.start()
    started.wait(10)
    yield f"http://127.0.0.1:{server.port}"
    loop.call_soon_threadsafe(loop.stop)
    discoverer.close()
    backend.stop(grace=None)


def test_get_initialize(gateway):
    r = httpx.get(gateway + "/")
    assert r.status_code == 200
    assert r.headers.get("mcp-session-id")
    assert r.json()["result"]["protocolVersion"] == "2024-11-05"


def test_tools_list_and_call_over_tcp(gateway):
    with httpx.Client(base_url=gateway) as client:
        r = client.post(
            "/", json={"jsonrpc": "2.0", "method": "tools/list", "id": 1}
        )
        names = [t["name"] for t in r.json()["result"]["tools"]]
        assert "hello_helloservice_sayhello" in names
        r = client.post(
            "/",
            json={
                "jsonrpc": "2.0",
                "method": "tools/call",
                "id": 2,
                "params": {
                    "name": "hello_helloservice_sayhello",
                    "arguments": {"name": "tcp"},
                },
            },
        )
        res = r.json()["result"]
        assert json.loads(res["content"][0]["text"]) == {"message": "Hello, tcp!"}


def test_health_and_metrics_over_tcp(gateway):
    assert httpx.get(gateway + "/health").status_code == 200
    m = httpx.get(gateway + "/metrics").json()
    assert m["methodCount"] >= 5


def test_keepalive_multiple_requests(gateway):
    with httpx.Client(base_url=gateway) as client:
        for i in range(5):
            r = client.post(
                "/", json={"jsonrpc": "2.0", "method": "initialize", "id": i}
            )
            assert r.status_code == 200


def test_404(gateway):
    assert httpx.get(gateway + "/nope").status_code == 404
