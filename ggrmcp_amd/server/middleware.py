"""HTTP middleware chain.

Re-design of the reference's ``pkg/server/middleware.go``: the same 10
middlewares — recovery (middleware.go:194-211), logging (17-43), security
headers (65-86), CORS with ``Mcp-Session-Id`` exposure (46-62), global
token-bucket rate limit 100 rps / burst 200 (89-102, 286), per-session rate
limit (105-130 — wired here, unlike the reference, and with a lock), content
type (133-161), body-size cap 1 MB (164-178), timeout (181-191) and metrics
(214-233 — a real counter here, not the reference's discard-the-duration
stub) — composed via ``chain_middleware`` (247-254) in the reference's
``DefaultMiddleware`` order (280-293).

Transport-agnostic: middlewares wrap ``async handler(Request) -> Response``;
both the asyncio HTTP frontend (server/http.py) and tests drive them
directly.
"""

from __future__ import annotations

import asyncio
import logging
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Awaitable, Callable, Dict, List, Optional

log = logging.getLogger("ggrmcp.http")


@dataclass
class Request:
    method: str = "GET"
    path: str = "/"
    headers: Dict[str, str] = field(default_factory=dict)  # lower-cased keys
    body: bytes = b""
    remote: str = ""


@dataclass
class Response:
    status: int = 200
    headers: Dict[str, str] = field(default_factory=dict)
    body: bytes = b""

    @classmethod
    def json(cls, payload: bytes, status: int = 200, **headers: str) -> "Response":
        h = {"Content-Type": "application/json"}
        h.update(headers)
        return cls(status=status, headers=h, body=payload)

    @classmethod
    def text(cls, message: str, status: int) -> "Response":
        return cls(status=status, headers={"Content-Type": "text/plain"}, body=message.encode())


Handler = Callable[[Request], Awaitable[Response]]
Middleware = Callable[[Handler], Handler]


def chain_middleware(handler: Handler, middlewares: List[Middleware]) -> Handler:
    """Reference ChainMiddleware (middleware.go:247-254): first in the list
    is outermost."""
    for mw in reversed(middlewares):
        handler = mw(handler)
    return handler


# -- individual middlewares ---------------------------------------------------


def recovery_middleware(next_h: Handler) -> Handler:
    """middleware.go:194-211."""

    async def handler(req: Request) -> Response:
        try:
            return await next_h(req)
        except asyncio.CancelledError:
            raise
        except Exception:
            log.exception("panic in handler")
            return Response.text("Internal Server Error", 500)

    return handler


def logging_middleware(next_h: Handler) -> Handler:
    """middleware.go:17-43."""

    async def handler(req: Request) -> Response:
        t0 = time.perf_counter()
        resp = await next_h(req)
        log.debug(
            "%s %s -> %d (%.2f ms)",
            req.method, req.path, resp.status, (time.perf_counter() - t0) * 1e3,
        )
        return resp

    return handler


def security_headers_middleware(next_h: Handler) -> Handler:
    """middleware.go:65-86 (CSP/HSTS and friends)."""

    async def handler(req: Request) -> Response:
        resp = await next_h(req)
        resp.headers.setdefault("X-Content-Type-Options", "nosniff")
        resp.headers.setdefault("X-Frame-Options", "DENY")
        resp.headers.setdefault("X-XSS-Protection", "1; mode=block")
        resp.headers.setdefault("Content-Security-Policy", "default-src 'none'")
        resp.headers.setdefault(
            "Strict-Transport-Security", "max-age=31536000; includeSubDomains"
        )
        return resp

    return handler


def cors_middleware(next_h: Handler) -> Handler:
    """middleware.go:46-62: permissive CORS, exposes Mcp-Session-Id."""

    async def handler(req: Request) -> Response:
        if req.method == "OPTIONS":
            resp = Response(status=204)
        else:
            resp = await next_h(req)
        resp.headers.setdefault("Access-Control-Allow-Origin", "*")
        resp.headers.setdefault("Access-Control-Allow-Methods", "GET, POST, OPTIONS")
        resp.headers.setdefault(
            "Access-Control-Allow-Headers", "Content-Type, Mcp-Session-Id, Authorization"
        )
        resp.headers.setdefault("Access-Control-Expose-Headers", "Mcp-Session-Id")
        return resp

    return handler


class TokenBucket:
    """Global token-bucket limiter (x/time/rate analog, middleware.go:89-102)."""

    def __init__(self, rate: float, burst: int) -> None:
        self.rate = rate
        self.burst = float(burst)
        self.tokens = float(burst)
        self.updated = time.monotonic()
        self._lock = threading.Lock()

    def allow(self, n: float = 1.0) -> bool:
        now = time.monotonic()
        with self._lock:
            self.tokens = min(self.burst, self.tokens + (now - self.updated) * self.rate)
            self.updated = now
            if self.tokens >= n:
                self.tokens -= n
                return True
            return False


def rate_limit_middleware(rate: float = 100.0, burst: int = 200) -> Middleware:
    """middleware.go:89-102 + 286."""
    bucket = TokenBucket(rate, burst)

    def mw(next_h: Handler) -> Handler:
        async def handler(req: Request) -> Response:
            if not bucket.allow():
                return Response.text("Too Many Requests", 429)
            return await next_h(req)

        return handler

    return mw


def content_type_middleware(next_h: Handler) -> Handler:
    """middleware.go:133-161: POST bodies must be JSON."""

    async def handler(req: Request) -> Response:
        if req.method == "POST":
            ctype = req.headers.get("content-type", "")
            if ctype and "application/json" not in ctype:
                return Response.text("Unsupported Media Type", 415)
        return await next_h(req)

    return handler


def max_bytes_middleware(limit: int = 1024 * 1024) -> Middleware:
    """middleware.go:164-178 (MaxBytesReader analog)."""

    def mw(next_h: Handler) -> Handler:
        async def handler(req: Request) -> Response:
            if len(req.body) > limit:
                return Response.text("Request Entity Too Large", 413)
            return await next_h(req)

        return handler

    return mw


def timeout_middleware(timeout_s: float = 30.0) -> Middleware:
    """middleware.go:181-191."""

    def mw(next_h: Handler) -> Handler:
        async def handler(req: Request) -> Response:
            try:
                return await asyncio.wait_for(next_h(req), timeout=timeout_s)
            except asyncio.TimeoutError:
                return Response.text("Request Timeout", 504)

        return handler

    return mw


class MetricsRecorder:
    """Counts requests/durations (a real version of middleware.go:214-233)."""

    def __init__(self) -> None:
        self.requests = 0
        self.errors = 0
        self.total_duration_s = 0.0
        self._lock = threading.Lock()

    def record(self, status: int, duration_s: float) -> None:
        with self._lock:
            self.requests += 1
            if status >= 500:
                self.errors += 1
            self.total_duration_s += duration_s

    def snapshot(self) -> Dict[str, Any]:
        with self._lock:
            avg = self.total_duration_s / self.requests if self.requests else 0.0
            return {
                "httpRequests": self.requests,
                "httpErrors": self.errors,
                "avgLatencyMs": avg * 1e3,
            }


def metrics_middleware(recorder: MetricsRecorder) -> Middleware:
    def mw(next_h: Handler) -> Handler:
        async def handler(req: Request) -> Response:
            t0 = time.perf_counter()
            resp = await next_h(req)
            recorder.record(resp.status, time.perf_counter() - t0)
            return resp

        return handler

    return mw


def default_middleware(
    cfg=None, recorder: Optional[MetricsRecorder] = None
) -> List[Middleware]:
    """Reference DefaultMiddleware order (middleware.go:280-293)."""
    from ..config import ServerConfig

    cfg = cfg or ServerConfig()
    mws: List[Middleware] = [recovery_middleware, logging_middleware]
    if cfg.security_headers_enabled:
        mws.append(security_headers_middleware)
    if cfg.cors_enabled:
        mws.append(cors_middleware)
    if cfg.rate_limit_enabled:
        mws.append(rate_limit_middleware(cfg.rate_limit_rps, cfg.rate_limit_burst))
    mws.append(content_type_middleware)
    mws.append(max_bytes_middleware(cfg.max_body_bytes))
    mws.append(timeout_middleware(cfg.handler_timeout_s))
    if recorder is not None:
        mws.append(metrics_middleware(recorder))
    return mws
