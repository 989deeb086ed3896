"""Asyncio HTTP/1.1 frontend.

The reference uses Go's ``net/http`` + gorilla/mux (cmd/grmcp/main.go:78-91,
202-208); here a small hand-rolled asyncio server provides the same surface —
``/`` (GET, POST, OPTIONS), ``/health``, ``/metrics`` — with keep-alive,
content-length bodies, the reference's read/write/idle timeouts and graceful
shutdown with a drain period (main.go:94-112).

This is the management/compat transport.  The high-throughput ingestion path
for benchmarks is the batch engine driven directly (bench.py), since a
dynamic-language TCP frontend would otherwise dominate the measurement.
"""

from __future__ import annotations

import asyncio
import logging
from typing import Callable, List, Optional

from .middleware import Handler, Middleware, Request, Response, chain_middleware

log = logging.getLogger("ggrmcp.httpserver")

_STATUS_TEXT = {
    200: "OK", 202: "Accepted", 204: "No Content", 400: "Bad Request",
    404: "Not Found", 405: "Method Not Allowed", 413: "Request Entity Too Large",
    415: "Unsupported Media Type", 429: "Too Many Requests",
    500: "Internal Server Error", 503: "Service Unavailable", 504: "Gateway Timeout",
}

MAX_HEADER_BYTES = 64 * 1024


class HTTPServer:
    def __init__(
        self,
        handler: Handler,
        middlewares: Optional[List[Middleware]] = None,
        host: str = "0.0.0.0",
        port: int = 50053,
        read_timeout_s: float = 15.0,
        idle_timeout_s: float = 60.0,
        max_body_bytes: int = 1024 * 1024,
    ) -> None:
        self.handler = chain_middleware(handler, middlewares or [])
        self.host = host
        self.port = port
        self.read_timeout_s = read_timeout_s
        self.idle_timeout_s = idle_timeout_s
        self.max_body_bytes = max_body_bytes
        self._server: Optional[asyncio.AbstractServer] = None

    async def start(self) -> None:
        self._server = await asyncio.start_server(
            self._serve_connection, self.host, self.port
        )
        addrs = ", ".join(str(s.getsockname()) for s in self._server.sockets)
        self.port = self._server.sockets[0].getsockname()[1]
        log.info("HTTP server listening on %s", addrs)

    async def stop(self, drain_s: float = 30.0) -> None:
        if self._server is not None:
            self._server.close()
            try:
                await asyncio.wait_for(self._server.wait_closed(), timeout=drain_s)
            except asyncio.TimeoutError:
                pass
            self._server = None

    async def serve_forever(self) -> None:
        if self._server is None:
            await self.start()
        async with self._server:
            await self._server.serve_forever()

    # -- connection handling ---------------------------------------------------

    async def _serve_connection(
        self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter
    ) -> None:
        peer = writer.get_extra_info("peername")
        remote = f"{peer[0]}:{peer[1]}" if isinstance(peer, tuple) else str(peer)
        try:
            first = True
            while True:
                timeout = self.read_timeout_s if first else self.idle_timeout_s
                try:
                    head = await asyncio.wait_for(
                        reader.readuntil(b"\r\n\r\n"), timeout=timeout
                    )
                except (
                    asyncio.TimeoutError,
                    asyncio.IncompleteReadError,
                    ConnectionResetError,
                ):
                    break
                except asyncio.LimitOverrunError:
                    await self._write_simple(writer, 400, close=True)
                    break
                first = False
                if len(head) > MAX_HEADER_BYTES:
                    await self._write_simple(writer, 400, close=True)
                    break
                req, keep_alive, err = self._parse_head(head, remote)
                if err is not None:
                    await self._write_simple(writer, err, close=True)
                    break
                clen = int(req.headers.get("content-length", "0") or 0)
                if clen > self.max_body_bytes:
                    await self._write_simple(writer, 413, close=True)
                    break
                if clen:
                    try:
                        req.body = await asyncio.wait_for(
                            reader.readexactly(clen), timeout=self.read_timeout_s
                        )
                    except (asyncio.TimeoutError, asyncio.IncompleteReadError):
                        break
                resp = await self.handler(req)
                await self._write_response(writer, resp, keep_alive)
                if not keep_alive:
                    break
        finally:
            try:
                writer.close()
                await writer.wait_closed()
            except Exception:
                pass

    @staticmethod
    def _parse_head(head: bytes, remote: str):
        try:
            text = head.decode("latin-1")
            lines = text.split("\r\n")
            method, path, version = lines[0].split(" ", 2)
        except ValueError:
            return None, False, 400
        headers = {}
        for line in lines[1:]:
            if not line:
                continue
            if ":" not in line:
                return None, False, 400
            k, v = line.split(":", 1)
            headers[k.strip().lower()] = v.strip()
        path = path.split("?", 1)[0]
        keep_alive = version.strip().endswith("1.1")
        conn = headers.get("connection", "").lower()
        if conn == "close":
            keep_alive = False
        elif conn == "keep-alive":
            keep_alive = True
        return Request(method=method, path=path, headers=headers, remote=remote), keep_alive, None

    async def _write_response(
        self, writer: asyncio.StreamWriter, resp: Response, keep_alive: bool
    ) -> None:
        status_text = _STATUS_TEXT.get(resp.status, "Unknown")
        parts = [f"HTTP/1.1 {resp.status} {status_text}\r\n"]
        headers = dict(resp.headers)
        headers["Content-Length"] = str(len(resp.body))
        headers["Connection"] = "keep-alive" if keep_alive else "close"
        for k, v in headers.items():
            parts.append(f"{k}: {v}\r\n")
        parts.append("\r\n")
        writer.write("".join(parts).encode("latin-1") + resp.body)
        await writer.drain()

    async def _write_simple(self, writer, status: int, close: bool = False) -> None:
        try:
            await self._write_response(writer, Response(status=status), not close)
        except Exception:
            pass
