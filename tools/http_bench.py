"""Serving-mode latency bench: per-request RTT through the REAL gateway.

Unlike bench.py (which drives GpuPipeline.process_batch with pre-built
batches), this spins the actual asyncio HTTP server + middleware +
MCPHandler + BatchEngineInvoker (GPU batch window) against a local backend,
then runs N concurrent MCP sessions each issuing sequential tools/call
requests over keep-alive TCP connections — the shape a real MCP deployment
sees.  Reports whole-gateway req/s and per-request p50/p90/p99 RTT.

  python tools/http_bench.py --sessions 256 --requests 50 [--no-gpu]

With --procs K the client side runs as K separate processes so the
measurement isn't capped by one client event loop.
"""

import argparse
import asyncio
import json
import random
import statistics
import sys
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))

from examples.bench_backend import serve_native  # noqa: E402
from ggrmcp_amd.cli import build_gateway  # noqa: E402
from ggrmcp_amd.config import Config  # noqa: E402
from ggrmcp_amd.server.http import HTTPServer  # noqa: E402
from ggrmcp_amd.server.middleware import MetricsRecorder, default_middleware  # noqa: E402
from ggrmcp_amd.utils.synthetic import hello_payload  # noqa: E402
from examples.protos import ALL_FDPS  # noqa: E402
from ggrmcp_amd.utils.synthetic import synthetic_fdp  # noqa: E402
from google.protobuf import descriptor_pb2  # noqa: E402


async def session_worker(port: int, sid: int, n_req: int, payload_bytes: int,
                         lat: list) -> None:
    rng = random.Random(sid)
    reader, writer = await asyncio.open_connection("127.0.0.1", port)
    try:
        for i in range(n_req):
            body = json.dumps({
                "jsonrpc": "2.0", "id": i, "method": "tools/call",
                "params": {"name": "hello_helloservice_sayhello",
                           "arguments": hello_payload(rng, payload_bytes)},
            }).encode()
            req = (
                b"POST / HTTP/1.1\r\nHost: l\r\nContent-Type: application/json\r\n"
                + f"Mcp-Session-Id: bench-{sid}\r\nContent-Length: {len(body)}\r\n\r\n".encode()
                + body
            )
            t0 = time.perf_counter()
            writer.write(req)
            await writer.drain()
            # read headers
            hdr = await reader.readuntil(b"\r\n\r\n")
            clen = 0
            for line in hdr.split(b"\r\n"):
                if line.lower().startswith(b"content-length:"):
                    clen = int(line.split(b":")[1])
            data = await reader.readexactly(clen)
            lat.append(time.perf_counter() - t0)
            resp = json.loads(data)
            assert resp.get("result", {}).get("isError") is False, resp
    finally:
        writer.close()


async def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--sessions", type=int, default=256)
    ap.add_argument("--requests", type=int, default=50)
    ap.add_argument("--payload-bytes", type=int, default=1024)
    ap.add_argument("--no-gpu", action="store_true")
    ap.add_argument("--native-frontend", action="store_true",
                    help="serve through the C++ reactor + GpuPipeline instead of asyncio")
    ap.add_argument("--batch-window-us", type=int, default=200)
    ap.add_argument("--procs", type=int, default=1, help="client processes")
    ap.add_argument("--streams", type=int, default=0,
                    help="override cfg.gpu.streams (0 = config default)")
    ap.add_argument("--cxx-client", action="store_true",
                    help="drive load with the C++ epoll client (no Python client ceiling)")
    args = ap.parse_args()

    srv, target = serve_native("127.0.0.1:0")
    cfg = Config.default()
    host, _, port = target.rpartition(":")
    cfg.grpc.host, cfg.grpc.port = host, int(port)
    cfg.gpu.enabled = not args.no_gpu
    cfg.gpu.batch_window_us = args.batch_window_us
    if args.streams > 0:
        cfg.gpu.streams = args.streams
    cfg.server.rate_limit_rps = 10_000_000  # measuring the engine, not the limiter
    cfg.server.rate_limit_burst = 10_000_000

    # native backend: no reflection; load the in-repo descriptor set
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer
    from ggrmcp_amd.cli import build_gateway as _bg  # noqa: F401

    # build_gateway does connect+discover; monkey-wire the descriptor path
    fdset = descriptor_pb2.FileDescriptorSet()
    fdset.file.extend(ALL_FDPS + [synthetic_fdp()])

    import ggrmcp_amd.backend.discovery as disc_mod

    orig_connect = disc_mod.ServiceDiscoverer.connect
    orig_discover = disc_mod.ServiceDiscoverer.discover

    def patched_connect(self, timeout_s=None):
        self.connections[0].connect(timeout_s=15)

    def patched_discover(self):
        self.load_descriptor_blob(fdset.SerializeToString())
        return self.tools

    disc_mod.ServiceDiscoverer.connect = patched_connect
    disc_mod.ServiceDiscoverer.discover = patched_discover
    try:
        handler, discoverer = build_gateway(cfg)
    finally:
        disc_mod.ServiceDiscoverer.connect = orig_connect
        disc_mod.ServiceDiscoverer.discover = orig_discover

    native_gw = None
    if args.native_frontend:
        from ggrmcp_amd.server.native_http import CpuBatchPipeline, NativeHTTPGateway

        cfg.server.rate_limit_rps = 10_000_000
        cfg.server.rate_limit_burst = 10_000_000
        pipe = (handler.invoker.pipeline if hasattr(handler.invoker, "pipeline")
                else CpuBatchPipeline(discoverer))
        native_gw = NativeHTTPGateway(pipe, discoverer, cfg)
        port_n = native_gw.start()

        class _FakeHTTP:
            port = port_n

            async def stop(self, *_a):
                native_gw.stop()

        http = _FakeHTTP()
    else:
        recorder = MetricsRecorder()
        http = HTTPServer(handler.handle,
                          middlewares=default_middleware(cfg.server, recorder),
                          port=0)
        await http.start()

    if args.cxx_client:
        from ggrmcp_amd.server.native_http import load_module as _lm

        femod = _lm()
        body = json.dumps({
            "jsonrpc": "2.0", "id": 1, "method": "tools/call",
            "params": {"name": "hello_helloservice_sayhello",
                       "arguments": hello_payload(random.Random(0), args.payload_bytes)},
        })
        femod.bench_client("127.0.0.1", http.port, min(args.sessions, 64), 3, body, 4)
        total, elapsed, pct, errs = femod.bench_client(
            "127.0.0.1", http.port, args.sessions, args.requests, body, 8)
        result = {
            "mode": "http-serving-native-cxxload",
            "sessions": args.sessions,
            "requests_per_session": args.requests,
            "payload_bytes": args.payload_bytes,
            "total_requests": total,
            "req_per_s": round(total / elapsed, 1),
            "p50_ms": round(pct[0] / 1e3, 3) if pct else None,
            "p90_ms": round(pct[1] / 1e3, 3) if pct else None,
            "p99_ms": round(pct[2] / 1e3, 3) if pct else None,
            "client_errors": errs,
        }
        print(json.dumps(result), flush=True)
        await http.stop(1.0)
        discoverer.close()
        srv.stop()
        return

    lat: list = []
    # warmup
    await asyncio.gather(*[
        session_worker(http.port, 10_000 + s, 3, args.payload_bytes, [])
        for s in range(min(args.sessions, 64))
    ])
    lat.clear()
    t0 = time.perf_counter()
    if args.procs > 1:
        import subprocess as sp

        per = args.sessions // args.procs
        procs = [
            sp.Popen([sys.executable, __file__, "--client-only",
                      str(http.port), str(k * per), str(per),
                      str(args.requests), str(args.payload_bytes)],
                     stdout=sp.PIPE, text=True)
            for k in range(args.procs)
        ]
        for p in procs:
            out, _ = p.communicate(timeout=600)
            lat.extend(json.loads(out.strip().splitlines()[-1]))
    else:
        await asyncio.gather(*[
            session_worker(http.port, s, args.requests, args.payload_bytes, lat)
            for s in range(args.sessions)
        ])
    dt = time.perf_counter() - t0

    lat.sort()
    n = len(lat)
    stats = getattr(handler.invoker, "stats", None)
    result = {
        "mode": "http-serving-native" if args.native_frontend else "http-serving",
        "gpu": cfg.gpu.enabled,
        "sessions": args.sessions,
        "requests_per_session": args.requests,
        "payload_bytes": args.payload_bytes,
        "total_requests": n,
        "req_per_s": round(n / dt, 1),
        "p50_ms": round(lat[n // 2] * 1e3, 3),
        "p90_ms": round(lat[int(n * 0.9)] * 1e3, 3),
        "p99_ms": round(lat[int(n * 0.99)] * 1e3, 3),
        "engine_stats": stats() if callable(stats) else None,
    }
    print(json.dumps(result), flush=True)
    await http.stop(1.0)
    discoverer.close()
    srv.stop()


async def client_only(argv) -> None:
    port, s0, count, n_req, payload = (int(x) for x in argv)
    lat: list = []
    await asyncio.gather(*[
        session_worker(port, s0 + s, n_req, payload, lat)
        for s in range(count)
    ])
    print(json.dumps(lat))


if __name__ == "__main__":
    if "--client-only" in sys.argv:
        i = sys.argv.index("--client-only")
        asyncio.run(client_only(sys.argv[i + 1 : i + 6]))
    else:
        asyncio.run(main())


# native C++ load generator entry (bypasses Python client limits):
#   python tools/http_bench.py --cxx-client --sessions 2048 --requests 50
