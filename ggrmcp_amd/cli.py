"""CLI bootstrap.

Re-design of the reference's ``cmd/grmcp/main.go``: the same flags
(``--grpc-host``, ``--grpc-port``, ``--http-port``, ``--log-level``,
``--dev``, ``--descriptor``; main.go:37-42) plus MI355X additions
(``--config`` file loading — the reference declares config files but never
loads them — ``--gpus``, ``--no-gpu``, ``--backends`` for centralized-gateway
mode).  Wires logger -> discoverer -> session manager -> tool builder ->
handler -> middleware -> HTTP server (main.go:137-199) with SIGINT/SIGTERM
graceful shutdown and a 30 s drain (main.go:94-112).

Run:  python -m ggrmcp_amd --grpc-host localhost --grpc-port 50051
"""

from __future__ import annotations

import argparse
import asyncio
import logging
import signal
import sys
from typing import List, Optional

from .backend.discovery import ServiceDiscoverer
from .config import Config, ConnectionConfig
from .headers import HeaderFilter
from .server.handler import CPUInvoker, MCPHandler
from .server.http import HTTPServer
from .server.middleware import MetricsRecorder, default_middleware
from .session import SessionManager
from .tools import MCPToolBuilder, build_comment_index

log = logging.getLogger("ggrmcp")


def parse_args(argv: Optional[List[str]] = None) -> argparse.Namespace:
    ap = argparse.ArgumentParser(prog="ggrmcp-amd", description=__doc__)
    ap.add_argument("--grpc-host", default=None, help="gRPC backend host (default localhost)")
    ap.add_argument("--grpc-port", type=int, default=None, help="gRPC backend port (default 50051)")
    ap.add_argument("--http-port", type=int, default=None, help="HTTP listen port (default 50053)")
    ap.add_argument("--log-level", default=None, choices=["debug", "info", "warn", "error"])
    ap.add_argument("--dev", action="store_true", help="development mode")
    ap.add_argument("--descriptor", default=None, help="FileDescriptorSet (.binpb) path")
    ap.add_argument("--config", default=None, help="YAML/JSON config file")
    ap.add_argument(
        "--backends",
        default=None,
        help="comma-separated extra host:port backends (centralized-gateway mode)",
    )
    ap.add_argument("--gpus", type=int, default=None, help="GPU engines for DP sharding")
    ap.add_argument("--no-gpu", action="store_true", help="CPU-only hot path")
    ap.add_argument(
        "--frontend",
        default="asyncio",
        choices=["asyncio", "native"],
        help="HTTP surface: asyncio (full middleware parity) or the C++ "
             "batch reactor (production serving path)",
    )
    return ap.parse_args(argv)


def build_config(args: argparse.Namespace) -> Config:
    if args.config:
        cfg = Config.from_file(args.config)
    elif args.dev:
        cfg = Config.development()
    else:
        cfg = Config.default()
    if args.grpc_host is not None:
        cfg.grpc.host = args.grpc_host
    if args.grpc_port is not None:
        cfg.grpc.port = args.grpc_port
    if args.http_port is not None:
        cfg.server.http_port = args.http_port
    if args.log_level is not None:
        cfg.logging.level = args.log_level
    if args.dev:
        cfg.logging.development = True
    if args.descriptor:
        cfg.descriptor_set.enabled = True
        cfg.descriptor_set.path = args.descriptor
    if args.backends:
        for spec in args.backends.split(","):
            host, _, port = spec.strip().rpartition(":")
            cfg.extra_backends.append(ConnectionConfig(host=host, port=int(port)))
    if args.gpus is not None:
        cfg.gpu.devices = args.gpus
    if args.no_gpu:
        cfg.gpu.enabled = False
    cfg.validate()
    return cfg


def setup_logging(cfg: Config) -> None:
    level = {"debug": logging.DEBUG, "info": logging.INFO, "warn": logging.WARNING,
             "warning": logging.WARNING, "error": logging.ERROR}[cfg.logging.level]
    fmt = (
        "%(asctime)s %(levelname)-5s %(name)s: %(message)s"
        if cfg.logging.development
        else '{"ts":"%(asctime)s","level":"%(levelname)s","logger":"%(name)s","msg":"%(message)s"}'
    )
    logging.basicConfig(level=level, format=fmt, stream=sys.stderr)


def build_gateway(cfg: Config):
    """Wire discoverer -> sessions -> tools -> handler (main.go:137-199).
    Returns (handler, discoverer)."""
    discoverer = ServiceDiscoverer(cfg)
    discoverer.connect()
    discoverer.discover()
    comment_index = build_comment_index(
        fdp for backend in discoverer._fdps for fdp in backend
    )
    sessions = SessionManager(
        ttl_s=cfg.session.ttl_s,
        cleanup_interval_s=cfg.session.cleanup_interval_s,
        max_sessions=cfg.session.max_sessions,
        rate_limit_per_min=cfg.session.rate_limit_per_min,
        rate_limit_burst=cfg.session.rate_limit_burst,
    )
    invoker = None
    if cfg.gpu.enabled:
        try:
            from .engine.batch import BatchEngineInvoker

            invoker = BatchEngineInvoker(discoverer, cfg)
        except Exception as e:
            if cfg.gpu.require_gpu:
                raise
            log.warning("GPU engine unavailable (%s); using CPU hot path", e)
    if invoker is None:
        invoker = CPUInvoker(discoverer)
    handler = MCPHandler(
        discoverer,
        session_manager=sessions,
        tool_builder=MCPToolBuilder(comment_index),
        header_filter=HeaderFilter.from_config(cfg.header_forwarding),
        config=cfg,
        invoker=invoker,
    )
    return handler, discoverer


async def run(cfg: Config, frontend: str = "asyncio") -> None:
    handler, discoverer = build_gateway(cfg)
    if frontend == "native":
        from .server.native_http import CpuBatchPipeline, NativeHTTPGateway

        if hasattr(handler.invoker, "pipeline"):
            pipeline = handler.invoker.pipeline
            if cfg.gpu.devices > 1:
                # one pipeline per GPU; sessions shard across them
                from .engine.batch import GpuPipeline

                pipeline = [pipeline] + [
                    GpuPipeline(discoverer, cfg, device=d)
                    for d in range(1, cfg.gpu.devices)
                ]
        else:
            pipeline = CpuBatchPipeline(discoverer)
        gw = NativeHTTPGateway(
            pipeline, discoverer, cfg,
            sessions=handler.sessions,
            tool_builder=handler.tool_builder,
            header_filter=handler.header_filter,
            host="0.0.0.0", port=cfg.server.http_port,
        )
        port = gw.start()
        stop = asyncio.Event()
        loop = asyncio.get_running_loop()
        for sig in (signal.SIGINT, signal.SIGTERM):
            try:
                loop.add_signal_handler(sig, stop.set)
            except NotImplementedError:  # pragma: no cover
                pass
        log.info("native gateway ready on :%d (backend %s)", port, cfg.grpc.target)
        await stop.wait()
        gw.stop()
        discoverer.close()
        return
    recorder = MetricsRecorder()
    server = HTTPServer(
        handler.handle,
        middlewares=default_middleware(cfg.server, recorder),
        port=cfg.server.http_port,
        read_timeout_s=cfg.server.read_timeout_s,
        idle_timeout_s=cfg.server.idle_timeout_s,
        max_body_bytes=cfg.server.max_body_bytes,
    )
    await server.start()

    stop = asyncio.Event()
    loop = asyncio.get_running_loop()
    for sig in (signal.SIGINT, signal.SIGTERM):
        try:
            loop.add_signal_handler(sig, stop.set)
        except NotImplementedError:  # pragma: no cover
            pass
    log.info("gateway ready on :%d (backend %s)", server.port, cfg.grpc.target)
    await stop.wait()
    log.info("shutting down (drain %.0fs)", cfg.server.shutdown_drain_s)
    await server.stop(cfg.server.shutdown_drain_s)
    discoverer.close()


def main(argv: Optional[List[str]] = None) -> None:
    args = parse_args(argv)
    cfg = build_config(args)
    setup_logging(cfg)
    asyncio.run(run(cfg, frontend=args.frontend))


if __name__ == "__main__":
    main()
