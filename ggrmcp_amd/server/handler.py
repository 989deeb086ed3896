"""MCP JSON-RPC protocol handler.

Re-design of the reference's ``pkg/server/handler.go``: GET -> initialize
result (handler.go:61-78); POST -> decode/validate/dispatch
(handler.go:81-139); ``initialize``, ``tools/list``, ``tools/call``,
``prompts/list``, ``resources/list`` (handler.go:142-287); JSON-RPC errors
always HTTP 200 (handler.go:311); gRPC call failures become
``ToolCallResult{isError:true}`` with HTTP 200 (handler.go:252-259); session
counters bumped per call (handler.go:262-263); ``/health`` 503 unless the
backend is healthy and at least one method is discovered (handler.go:331-364);
``/metrics`` returns the service stats JSON (handler.go:367-376).

The invocation seam is async: the handler awaits an *invoker* — either the
CPU reference path (``CPUInvoker`` wrapping
ServiceDiscoverer.invoke_method_by_tool in a thread pool) or the MI355X batch
engine (ggrmcp_amd/engine), which collects concurrent calls into GPU batches.
Extensions over the reference: JSON-RPC notifications (no id) are accepted
with 202, and server-streaming tools return one text content block per
message (the reference rejects streaming outright, discovery.go:354-356).
"""

from __future__ import annotations

import asyncio
import json
import logging
import time
from concurrent.futures import ThreadPoolExecutor
from typing import Any, Dict, Optional, Tuple

import grpc

from ..backend.discovery import (
    STREAMING_UNSUPPORTED_MSG,
    MethodNotFoundError,
    ServiceDiscoverer,
)
from ..config import Config
from ..headers import HeaderFilter
from ..mcp import types as mcp
from ..mcp.validation import ValidationError, Validator, sanitize_error
from ..session import SessionManager
from ..tools import MCPToolBuilder
from .middleware import Request, Response

log = logging.getLogger("ggrmcp.handler")

SESSION_HEADER = "mcp-session-id"


class CPUInvoker:
    """Host-side reference invoker (the reference's only path)."""

    def __init__(self, discoverer: ServiceDiscoverer, max_workers: int = 64) -> None:
        self.discoverer = discoverer
        self._pool = ThreadPoolExecutor(max_workers=max_workers, thread_name_prefix="invoke")

    async def invoke(
        self, tool_name: str, args_json: str, headers: Dict[str, str], timeout_s: float
    ) -> str:
        loop = asyncio.get_running_loop()
        return await loop.run_in_executor(
            self._pool,
            lambda: self.discoverer.invoke_method_by_tool(
                tool_name, args_json, headers, timeout_s
            ),
        )

    async def invoke_streaming(
        self, tool_name: str, args_json: str, headers: Dict[str, str], timeout_s: float
    ):
        loop = asyncio.get_running_loop()

        def collect():
            return list(
                self.discoverer.invoke_streaming(tool_name, args_json, headers, timeout_s)
            )

        return await loop.run_in_executor(self._pool, collect)

    def close(self) -> None:
        self._pool.shutdown(wait=False)


class MCPHandler:
    """Reference server.Handler."""

    def __init__(
        self,
        discoverer: ServiceDiscoverer,
        session_manager: Optional[SessionManager] = None,
        tool_builder: Optional[MCPToolBuilder] = None,
        header_filter: Optional[HeaderFilter] = None,
        config: Optional[Config] = None,
        invoker=None,
    ) -> None:
        self.config = config or Config.default()
        self.discoverer = discoverer
        self.sessions = session_manager or SessionManager(
            ttl_s=self.config.session.ttl_s,
            cleanup_interval_s=self.config.session.cleanup_interval_s,
            max_sessions=self.config.session.max_sessions,
            rate_limit_per_min=self.config.session.rate_limit_per_min,
            rate_limit_burst=self.config.session.rate_limit_burst,
        )
        self.tool_builder = tool_builder or MCPToolBuilder()
        self.header_filter = header_filter or HeaderFilter.from_config(
            self.config.header_forwarding
        )
        self.validator = Validator()
        self.invoker = invoker or CPUInvoker(discoverer)
        self.start_time = time.time()

    # -- top-level routing ----------------------------------------------------

    async def handle(self, req: Request) -> Response:
        if req.path == "/health":
            return await self.handle_health(req)
        if req.path == "/metrics":
            return self.handle_metrics(req)
        if req.path != "/":
            return Response.text("Not Found", 404)
        if req.method == "GET":
            return self.handle_get(req)
        if req.method == "POST":
            return await self.handle_post(req)
        if req.method == "OPTIONS":
            return Response(status=204)
        return Response.text("Method Not Allowed", 405)

    # -- GET / (handler.go:61-78) ---------------------------------------------

    def handle_get(self, req: Request) -> Response:
        session = self.sessions.get_or_create(req.headers.get(SESSION_HEADER), req.headers)
        resp = mcp.JSONRPCResponse(id=None, result=mcp.initialization_result())
        return Response.json(
            json.dumps(resp.to_dict()).encode(), **{"Mcp-Session-Id": session.id}
        )

    # -- POST / (handler.go:81-139) ---------------------------------------------

    async def handle_post(self, req: Request) -> Response:
        try:
            data = json.loads(req.body or b"{}")
        except (json.JSONDecodeError, UnicodeDecodeError) as e:
            return self._error_response(None, mcp.PARSE_ERROR, f"parse error: {e}")
        if not isinstance(data, dict):
            return self._error_response(None, mcp.INVALID_REQUEST, "request must be an object")

        rid = data.get("id")
        method = data.get("method")
        # notifications (no id): accept and drop (MCP clients send
        # notifications/initialized; extension over the reference)
        if "id" not in data and isinstance(method, str) and method.startswith("notifications/"):
            return Response(status=202)

        try:
            self.validator.validate_request(data)
        except ValidationError as e:
            return self._error_response(rid, mcp.INVALID_REQUEST, str(e))

        session = self.sessions.get_or_create(req.headers.get(SESSION_HEADER), req.headers)
        request = mcp.JSONRPCRequest.from_dict(data)
        try:
            result, status = await self.handle_request(request, session)
        except mcp.JSONRPCError as e:
            return self._error_response(
                rid, e.error.code, e.error.message, e.error.data, session_id=session.id
            )
        except ValidationError as e:
            return self._error_response(rid, mcp.INVALID_PARAMS, str(e), session_id=session.id)
        except MethodNotFoundError as e:
            return self._error_response(
                rid, mcp.METHOD_NOT_FOUND, str(e.args[0]), session_id=session.id
            )
        except Exception as e:  # error-code mapping by kind (handler.go:117-127)
            log.exception("internal error handling %s", request.method)
            return self._error_response(
                rid, mcp.INTERNAL_ERROR, sanitize_error(str(e)), session_id=session.id
            )
        resp = mcp.JSONRPCResponse(id=rid, result=result)
        return Response.json(
            json.dumps(resp.to_dict(), ensure_ascii=False).encode(),
            status=status,
            **{"Mcp-Session-Id": session.id},
        )

    # -- dispatch (handler.go:142-156) -----------------------------------------

    async def handle_request(self, request: mcp.JSONRPCRequest, session) -> Tuple[Any, int]:
        method = request.method
        if method == "initialize":
            return mcp.initialization_result(), 200
        if method == "notifications/initialized":
            return {}, 200
        if method == "tools/list":
            return self.handle_tools_list(), 200
        if method == "tools/call":
            return await self.handle_tools_call(request, session), 200
        if method == "prompts/list":
            return {"prompts": []}, 200  # handler.go:274-287
        if method == "resources/list":
            return {"resources": []}, 200
        if method == "ping":
            return {}, 200
        raise MethodNotFoundError(f"method not found: {method}")

    # -- tools/list (handler.go:182-212) ----------------------------------------

    def handle_tools_list(self) -> Dict[str, Any]:
        self.tool_builder.set_cache_key(self.discoverer.tools_version)
        tools = self.tool_builder.build_tools(self.discoverer.get_methods())
        return {"tools": [t.to_dict() for t in tools]}

    # -- tools/call (handler.go:215-271) ------------------------------------------

    async def handle_tools_call(self, request: mcp.JSONRPCRequest, session) -> Dict[str, Any]:
        tool_name = self.validator.validate_tool_call_params(request.params)
        if session.is_blocked:
            raise mcp.JSONRPCError(mcp.INVALID_REQUEST, "session is blocked")
        if self.config.session.rate_limit_enabled and not self.sessions.check_rate_limit(
            session
        ):
            raise mcp.JSONRPCError(mcp.INVALID_REQUEST, "session rate limit exceeded")

        args = (request.params or {}).get("arguments")
        args_json = json.dumps(args if args is not None else {}, ensure_ascii=False)
        headers = self.header_filter.filter_headers(session.headers)
        timeout_s = self.config.server.handler_timeout_s

        mi = self.discoverer.get_method_by_tool(tool_name)  # raises MethodNotFound
        try:
            if mi.is_server_streaming:
                chunks = await self.invoker.invoke_streaming(
                    tool_name, args_json, headers, timeout_s
                )
                content = [mcp.TextContent(c) for c in chunks]
            else:
                output = await self.invoker.invoke(tool_name, args_json, headers, timeout_s)
                content = [mcp.TextContent(output)]
            result = mcp.ToolCallResult(content=content, is_error=False)
        except grpc.RpcError as e:
            # gRPC errors -> isError result, HTTP 200 (handler.go:252-259)
            code = e.code().name if hasattr(e, "code") else "UNKNOWN"
            detail = e.details() if hasattr(e, "details") else str(e)
            result = mcp.ToolCallResult(
                content=[mcp.TextContent(f"gRPC error {code}: {sanitize_error(detail)}")],
                is_error=True,
            )
        except ValueError as e:
            if STREAMING_UNSUPPORTED_MSG in str(e):
                raise mcp.JSONRPCError(mcp.INVALID_PARAMS, str(e))
            result = mcp.ToolCallResult(
                content=[mcp.TextContent(sanitize_error(str(e)))], is_error=True
            )
        session.increment_call_count()
        session.update_last_accessed()
        return result.to_dict()

    # -- /health (handler.go:331-364) ---------------------------------------------

    async def handle_health(self, req: Request) -> Response:
        loop = asyncio.get_running_loop()
        healthy = await loop.run_in_executor(None, self.discoverer.health_check)
        method_count = len(self.discoverer.tools)
        ok = healthy and method_count > 0
        payload = {
            "status": "healthy" if ok else "unhealthy",
            "timestamp": time.time(),
            "serviceCount": self.discoverer.stats()["serviceCount"],
            "methodCount": method_count,
        }
        return Response.json(json.dumps(payload).encode(), status=200 if ok else 503)

    # -- /metrics (handler.go:367-376) ----------------------------------------------

    def handle_metrics(self, req: Request) -> Response:
        stats = self.discoverer.stats()
        stats["sessions"] = self.sessions.stats()
        stats["uptimeS"] = time.time() - self.start_time
        engine = getattr(self.invoker, "stats", None)
        if callable(engine):
            stats["engine"] = engine()
        return Response.json(json.dumps(stats).encode())

    # -- helpers ---------------------------------------------------------------------

    def _error_response(
        self,
        rid,
        code: int,
        message: str,
        data: Any = None,
        session_id: Optional[str] = None,
    ) -> Response:
        resp = mcp.JSONRPCResponse(id=rid, error=mcp.RPCError(code, message, data))
        headers = {"Mcp-Session-Id": session_id} if session_id else {}
        # JSON-RPC errors are HTTP 200 (handler.go:311)
        return Response.json(json.dumps(resp.to_dict()).encode(), **headers)
