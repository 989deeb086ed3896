"""Demo gRPC backend with reflection — the reference's
``examples/hello-service`` (main.go:20-58) re-built for this environment.

Serves hello.HelloService plus the complex services over TCP or a Unix
socket, with gRPC server reflection (v1 + v1alpha) enabled.  Run directly:

    python examples/hello_service.py --port 50051
"""

from __future__ import annotations

import argparse
import sys
import time
from concurrent import futures
from pathlib import Path

import grpc

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from examples.protos import ALL_FDPS, SERVICE_NAMES  # noqa: E402
from ggrmcp_amd.backend.reflection_server import enable_reflection  # noqa: E402
from ggrmcp_amd.utils.protobuild import message_class, pool_for  # noqa: E402

POOL = pool_for(ALL_FDPS)
HelloRequest = message_class(POOL, "hello.HelloRequest")
HelloResponse = message_class(POOL, "hello.HelloResponse")
GetUserRequest = message_class(POOL, "complex.GetUserRequest")
UserProfile = message_class(POOL, "complex.UserProfile")
Document = message_class(POOL, "complex.Document")
NodeRequest = message_class(POOL, "complex.NodeRequest")


def _say_hello(request, context):
    # "error" input triggers a gRPC error — mirrors the reference test
    # backend's error-path triggers (tests/test_utils.go:229-231).
    if request.name == "error":
        context.abort(grpc.StatusCode.INVALID_ARGUMENT, "name must not be 'error'")
    return HelloResponse(message=f"Hello, {request.name}!")


def _get_user(request, context):
    if not request.user_id:
        context.abort(grpc.StatusCode.NOT_FOUND, "user not found")
    profile = UserProfile(
        user_id=request.user_id,
        name=f"User {request.user_id}",
        status=1,  # STATUS_ACTIVE
        tags=["alpha", "beta"],
        score=9_007_199_254_740_993,  # > 2^53: exercises int64-as-string JSON
        rating=4.5,
        avatar=b"\x00\x01\x02",
    )
    profile.created_at.seconds = 1_700_000_000
    profile.created_at.nanos = 123_000_000
    return profile


def _put_document(request, context):
    return request  # echo (oneof + map round trip)


def _echo_node(request, context):
    return request  # echo (recursive message round trip)


def _stream_nodes(request, context):
    n = max(1, request.depth or 1)
    for i in range(n):
        out = NodeRequest()
        out.CopyFrom(request)
        out.depth = i
        yield out


def _unary(handler, req_cls):
    return grpc.unary_unary_rpc_method_handler(
        handler,
        request_deserializer=req_cls.FromString,
        response_serializer=lambda m: m.SerializeToString(),
    )


def _server_stream(handler, req_cls):
    return grpc.unary_stream_rpc_method_handler(
        handler,
        request_deserializer=req_cls.FromString,
        response_serializer=lambda m: m.SerializeToString(),
    )


def build_server(max_workers: int = 16) -> grpc.Server:
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers))
    server.add_generic_rpc_handlers(
        (
            grpc.method_handlers_generic_handler(
                "hello.HelloService", {"SayHello": _unary(_say_hello, HelloRequest)}
            ),
            grpc.method_handlers_generic_handler(
                "complex.UserService", {"GetUser": _unary(_get_user, GetUserRequest)}
            ),
            grpc.method_handlers_generic_handler(
                "complex.DocumentService",
                {"PutDocument": _unary(_put_document, Document)},
            ),
            grpc.method_handlers_generic_handler(
                "complex.NodeService",
                {
                    "Echo": _unary(_echo_node, NodeRequest),
                    "StreamNodes": _server_stream(_stream_nodes, NodeRequest),
                },
            ),
        )
    )
    enable_reflection(server, SERVICE_NAMES, ALL_FDPS)
    return server


def serve(target: str = "127.0.0.1:0", max_workers: int = 16):
    """Start the demo backend; returns (server, bound_target)."""
    server = build_server(max_workers)
    if target.startswith("unix:"):
        server.add_insecure_port(target)
        bound = target
    else:
        host, _, port = target.rpartition(":")
        actual = server.add_insecure_port(target)
        bound = f"{host}:{actual}"
    server.start()
    return server, bound


def main() -> None:
    ap = argparse.ArgumentParser(description="ggrmcp-amd demo gRPC backend")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=50051)
    ap.add_argument("--uds", default="", help="serve on unix:PATH instead of TCP")
    args = ap.parse_args()
    target = f"unix:{args.uds}" if args.uds else f"{args.host}:{args.port}"
    server, bound = serve(target)
    print(f"hello-service listening on {bound}", flush=True)
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        server.stop(grace=2)


if __name__ == "__main__":
    main()
