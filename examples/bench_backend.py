"""Benchmark backend: the demo services plus the synthetic bench.EchoService
(BASELINE config 3's 64-field proto).  Runs as a separate process so gateway
and backend don't share a GIL.

    python -m examples.bench_backend --uds /tmp/bench.sock
"""

from __future__ import annotations

import argparse
import sys
import time
from pathlib import Path

import grpc

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from examples.hello_service import build_server  # noqa: E402
from examples.protos import ALL_FDPS, SERVICE_NAMES  # noqa: E402
from ggrmcp_amd.backend.reflection_server import ReflectionServicer  # noqa: E402
from ggrmcp_amd.utils.protobuild import message_class, pool_for  # noqa: E402
from ggrmcp_amd.utils.synthetic import synthetic_fdp  # noqa: E402

BENCH_FDP = synthetic_fdp()
ALL = ALL_FDPS + [BENCH_FDP]
NAMES = SERVICE_NAMES + ["bench.EchoService"]


def build_bench_server(max_workers: int = 32, package: str = "bench") -> grpc.Server:
    """Serve ``package``.EchoService (Echo unary + StreamEcho server-stream)
    plus the demo services.  Distinct packages distinguish backends in
    centralized-gateway mode (BASELINE config 5)."""
    from concurrent import futures

    fdps = ALL if package == "bench" else ALL_FDPS + [synthetic_fdp(package=package)]
    pool = pool_for(fdps)
    Wide64 = message_class(pool, f"{package}.Wide64")

    def _echo(request, context):
        return request

    def _stream_echo(request_bytes, context):
        """Raw-bytes streaming echo: f02_int32 = requested message count
        (BASELINE config 4: 4096 msgs/stream).  Bytes in, bytes out — the
        per-chunk serialize cost stays off the bench's critical path."""
        msg = Wide64.FromString(request_bytes)
        n = min(max(msg.f02_int32, 1), 1 << 16)
        for _ in range(n):
            yield request_bytes

    stream_echo_handler = grpc.unary_stream_rpc_method_handler(
        _stream_echo,
        request_deserializer=lambda b: b,
        response_serializer=lambda b: b,
    )

    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers))
    # demo services (rebuild handlers against this server)
    import examples.hello_service as hs

    server.add_generic_rpc_handlers(
        (
            grpc.method_handlers_generic_handler(
                "hello.HelloService",
                {"SayHello": hs._unary(hs._say_hello, hs.HelloRequest)},
            ),
            grpc.method_handlers_generic_handler(
                "complex.UserService",
                {"GetUser": hs._unary(hs._get_user, hs.GetUserRequest)},
            ),
            grpc.method_handlers_generic_handler(
                "complex.DocumentService",
                {"PutDocument": hs._unary(hs._put_document, hs.Document)},
            ),
            grpc.method_handlers_generic_handler(
                "complex.NodeService",
                {
                    "Echo": hs._unary(hs._echo_node, hs.NodeRequest),
                    "StreamNodes": hs._server_stream(hs._stream_nodes, hs.NodeRequest),
                },
            ),
            grpc.method_handlers_generic_handler(
                f"{package}.EchoService",
                {"Echo": hs._unary(_echo, Wide64), "StreamEcho": stream_echo_handler},
            ),
        )
    )
    names = SERVICE_NAMES + [f"{package}.EchoService"]
    servicer = ReflectionServicer(names, fdps)
    server.add_generic_rpc_handlers(tuple(servicer.generic_handlers()))
    return server


def serve(target: str = "127.0.0.1:0", max_workers: int = 32, package: str = "bench"):
    server = build_bench_server(max_workers, package)
    if target.startswith("unix:"):
        server.add_insecure_port(target)
        bound = target
    else:
        host, _, port = target.rpartition(":")
        actual = server.add_insecure_port(target)
        bound = f"{host}:{actual}"
    server.start()
    return server, bound


def serve_native(target: str, package: str = "bench"):
    """C++ nghttp2 backend (ops/csrc/h2grpc.cpp H2Server): native handlers
    for SayHello, unary echo and server-streaming echo routes; no Python in
    the serving path."""
    from ggrmcp_amd.backend.native_invoker import load_module

    mod = load_module()
    srv = mod.Server(target)
    srv.add_route("/hello.HelloService/SayHello", "hello")
    srv.add_route(f"/{package}.EchoService/Echo", "echo")
    srv.add_route(f"/{package}.EchoService/StreamEcho", "stream_echo")
    srv.add_route("/complex.DocumentService/PutDocument", "echo")
    srv.add_route("/complex.NodeService/Echo", "echo")
    bound = srv.start()
    return srv, bound


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=0)
    ap.add_argument("--uds", default="")
    ap.add_argument("--workers", type=int, default=32)
    ap.add_argument("--native", action="store_true",
                    help="serve with the C++ nghttp2 backend instead of grpcio")
    ap.add_argument("--package", default="bench",
                    help="synthetic EchoService package (centralized-gateway mode)")
    args = ap.parse_args()
    target = f"unix:{args.uds}" if args.uds else f"{args.host}:{args.port}"
    if args.native:
        server, bound = serve_native(target, args.package)
        print(f"READY {bound}", flush=True)
        try:
            while True:
                time.sleep(3600)
        except KeyboardInterrupt:
            server.stop()
        return
    server, bound = serve(target, args.workers, args.package)
    print(f"READY {bound}", flush=True)
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        server.stop(grace=1)


if __name__ == "__main__":
    main()
