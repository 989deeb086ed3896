"""Native serving gateway: the C++ ingestion front end + the GPU pipeline.

The asyncio surface (server/http.py) provides full middleware/semantic
parity with the reference; this module is the MI355X serving path for
production throughput: ops/csrc/frontend.cpp accepts HTTP/1.1 MCP traffic,
batches concurrent POST bodies over an adaptive window, and calls
``batch_cb`` ONCE per batch — which runs the whole GPU hot path
(GpuPipeline.process_batch).  JSON-RPC envelopes that are not ``tools/call``
(initialize / tools/list / prompts / resources) are flagged by the encode
kernel (E_NOT_TOOLCALL) and answered synchronously by the same MCP logic the
asyncio handler uses; GET /, /health, /metrics and OPTIONS take the slow
callback one at a time.
"""

from __future__ import annotations

import importlib
import json
import logging
import sys
import threading
import time
from pathlib import Path
from typing import Dict, List, Optional, Tuple

from ..config import Config
from ..headers import HeaderFilter
from ..mcp import types as mcp
from ..mcp.validation import Validator
from ..session import SessionManager
from ..tools import MCPToolBuilder

log = logging.getLogger("ggrmcp.native_http")


class CpuBatchPipeline:
    """CPU fallback pipeline for the native front end (no GPU present):
    the reference-equivalent per-request path, fanned out over a thread
    pool, with the same batch/envelope contract as GpuPipeline."""

    def __init__(self, discoverer, invoke_workers: int = 64) -> None:
        from concurrent.futures import ThreadPoolExecutor

        self.discoverer = discoverer
        self.non_toolcall_handler = None
        self._pool = ThreadPoolExecutor(max_workers=invoke_workers,
                                        thread_name_prefix="cpubatch")

        class _S:
            def snapshot(self):
                return {"mode": "cpu"}

        class _E:
            stats = _S()

        self.engine = _E()

    def process_batch(self, bodies, headers=None, timeout_s=None):
        def one(i):
            body = bodies[i]
            hdr = headers[i] if headers else None
            try:
                data = json.loads(body)
            except Exception:
                resp = mcp.JSONRPCResponse(
                    id=None, error=mcp.RPCError(mcp.PARSE_ERROR, "parse error"))
                return json.dumps(resp.to_dict()).encode()
            if data.get("method") != "tools/call":
                if self.non_toolcall_handler is not None:
                    return self.non_toolcall_handler(body, hdr)
                resp = mcp.JSONRPCResponse(
                    id=data.get("id"),
                    error=mcp.RPCError(mcp.METHOD_NOT_FOUND, "method not found"))
                return json.dumps(resp.to_dict()).encode()
            rid = data.get("id")
            params = data.get("params") or {}
            try:
                mi = self.discoverer.get_method_by_tool(params.get("name", ""))
                args_json = json.dumps(params.get("arguments", {}), ensure_ascii=False)
                if mi.is_server_streaming:
                    chunks = list(self.discoverer.invoke_streaming(
                        params["name"], args_json, hdr, timeout_s))
                    result = mcp.ToolCallResult(
                        content=[mcp.TextContent(c) for c in chunks], is_error=False)
                else:
                    out = self.discoverer.invoke_method_by_tool(
                        params.get("name", ""), args_json, hdr, timeout_s)
                    result = mcp.ToolCallResult(
                        content=[mcp.TextContent(out)], is_error=False)
                resp = mcp.JSONRPCResponse(id=rid, result=result.to_dict())
            except KeyError:
                resp = mcp.JSONRPCResponse(
                    id=rid, error=mcp.RPCError(mcp.METHOD_NOT_FOUND, "tool not found"))
            except Exception as e:
                resp = mcp.JSONRPCResponse(
                    id=rid, error=mcp.RPCError(mcp.INTERNAL_ERROR, str(e)[:256]))
            return json.dumps(resp.to_dict(), ensure_ascii=False).encode()

        return list(self._pool.map(one, range(len(bodies))))


def load_module():
    ops_dir = str(Path(__file__).resolve().parent.parent / "ops")
    if ops_dir not in sys.path:
        sys.path.insert(0, ops_dir)
    return importlib.import_module("_frontend")


class _SidContext:
    """Minimal SessionContext stand-in for slow paths that only need .id."""

    __slots__ = ("id",)

    def __init__(self, sid: str) -> None:
        self.id = sid


class NativeSessionStore:
    """SessionManager facade over the C++ shared-memory SessionTable
    (ops/csrc/session_table.h; reference semantics pkg/session/manager.go).

    The serving hot path never calls into this — the guard runs inside the
    C++ reactor at parse time.  This facade gives the slow paths (GET /
    session issuance, /metrics, admin block/unblock) the same view of the
    same state.  Divergence from the Python SessionManager: the table keeps
    counters/flags only, not header snapshots (headers travel per-request
    on this path), so session_info reports headerCount 0."""

    def __init__(self, table) -> None:
        self.table = table

    def guard(self, session_id, headers=None, rate_limit: bool = True):
        sid, verdict, _created = self.table.guard(session_id or "", rate_limit)
        return _SidContext(sid), verdict

    def get_or_create(self, session_id, headers=None) -> _SidContext:
        return _SidContext(self.table.get_or_create(session_id or ""))

    def block(self, session_id: str) -> bool:
        return self.table.block(session_id)

    def unblock(self, session_id: str) -> bool:
        return self.table.unblock(session_id)

    def remove(self, session_id: str) -> bool:
        return self.table.remove(session_id)

    def session_info(self, session_id: str):
        info = self.table.info(session_id)
        if info is not None:
            info["headerCount"] = 0
        return info

    def stats(self):
        return self.table.stats()


class NativeHTTPGateway:
    """Drop-in serving front end over one or more GpuPipelines.

    With several pipelines (one per GPU device), sessions shard stably
    across them by ``Mcp-Session-Id`` ("sessions hashed to 1
    of 8 GPUs; per-GPU session tables"): each batch splits into per-shard
    sub-batches that run concurrently, so every GPU owns its sessions
    end-to-end and no request data crosses devices."""

    def __init__(
        self,
        pipeline,
        discoverer,
        config: Optional[Config] = None,
        sessions: Optional[SessionManager] = None,
        tool_builder: Optional[MCPToolBuilder] = None,
        header_filter: Optional[HeaderFilter] = None,
        host: str = "127.0.0.1",
        port: int = 0,
    ) -> None:
        self.config = config or Config.default()
        self.pipelines = list(pipeline) if isinstance(pipeline, (list, tuple)) else [pipeline]
        self.pipeline = self.pipelines[0]
        self.discoverer = discoverer
        self.tools = tool_builder or MCPToolBuilder()
        self.headers = header_filter or HeaderFilter.from_config(
            self.config.header_forwarding
        )
        self.validator = Validator()
        self.start_time = time.time()
        for p in self.pipelines:
            p.non_toolcall_handler = self._handle_non_toolcall
        if len(self.pipelines) > 1:
            from concurrent.futures import ThreadPoolExecutor

            self._shard_pool = ThreadPoolExecutor(
                max_workers=len(self.pipelines), thread_name_prefix="gshard"
            )
        else:
            self._shard_pool = None
        mod = load_module()
        srv_cfg = self.config.server
        sess_cfg = self.config.session
        # session state: C++ shared-memory table (reactor-side guard, shared
        # across serve_dp ranks when shared_table_path is set) unless the
        # caller injected a Python SessionManager explicitly
        self._cxx_sessions = sessions is None and hasattr(mod, "SessionTable")
        if self._cxx_sessions:
            self._sess_table = mod.SessionTable(
                capacity=max(1024, 2 * sess_cfg.max_sessions),
                ttl_s=sess_cfg.ttl_s,
                path=sess_cfg.shared_table_path,
                rate_per_min=sess_cfg.rate_limit_per_min,
                rate_burst=sess_cfg.rate_limit_burst,
            )
            self.sessions = NativeSessionStore(self._sess_table)
        else:
            self._sess_table = None
            self.sessions = sessions or SessionManager(
                ttl_s=sess_cfg.ttl_s,
                max_sessions=sess_cfg.max_sessions,
                rate_limit_per_min=sess_cfg.rate_limit_per_min,
                rate_limit_burst=sess_cfg.rate_limit_burst,
            )
        # case-insensitive filters (the default) run inside the C++ parser:
        # the batch callback then receives pre-filtered lowercase names and
        # skips the per-request Python dict rebuild.  Case-sensitive configs
        # keep the Python filter (C++ normalizes to lowercase).
        self._cxx_header_filter = bool(
            getattr(self.headers, "case_insensitive", False)
            and hasattr(mod.Frontend, "set_header_filter")
        )
        # fully-native span: the frontend workers call the GPU engine's
        # ISpanExecutor directly (no GIL on the tools/call hot path).
        # Requires the C++ session guard + header filter (so no per-request
        # Python is needed) and a single GPU pipeline with native transport.
        native_span_ready = (
            self._cxx_sessions
            and self._cxx_header_filter
            and len(self.pipelines) == 1
            and getattr(self.pipeline, "wire_clients", None)
            and hasattr(self.pipeline, "engines")
            and hasattr(mod.Frontend, "set_native_span")
        )
        n_workers = max(1, self.config.gpu.streams) if native_span_ready else 1
        self._fe = mod.Frontend(
            host,
            port,
            self._batch_cb,
            self._slow_cb,
            batch_window_us=self.config.gpu.batch_window_us,
            max_batch=self.config.gpu.max_batch,
            max_body=srv_cfg.max_body_bytes,
            rate_rps=float(srv_cfg.rate_limit_rps),
            rate_burst=float(srv_cfg.rate_limit_burst),
            workers=n_workers,
            reactors=max(1, getattr(srv_cfg, "reactors", 4)),
        )
        if getattr(srv_cfg, "reuse_port", False) and hasattr(self._fe, "set_reuse_port"):
            self._fe.set_reuse_port(True)
        if self._sess_table is not None:
            self._fe.set_session_table(
                self._sess_table, sess_cfg.rate_limit_enabled
            )
        self._span_engines: List = []
        if native_span_ready:
            try:
                self._setup_native_span()
            except Exception as e:  # pragma: no cover - env-specific
                log.warning("native span unavailable (%s); using batch_cb", e)
        if self._cxx_header_filter:
            self._fe.set_header_filter(
                self.headers.enabled,
                self.headers.forward_all,
                sorted(self.headers.allowed),
                sorted(self.headers.blocked),
            )
        self.port = 0

    def _setup_native_span(self) -> None:
        """Dedicated span engines for the C++ workers (separate from the
        Python pipeline's engines so worker-held arenas and Python calls
        never interleave on one engine; HBM3E has room to spare)."""
        from ..engine.batch import GpuEngine

        cfg = self.config
        n = max(1, cfg.gpu.streams)
        tables = self.pipeline.engine.tables
        stats = self.pipeline.engine.stats
        self._span_engines = [
            GpuEngine(self.discoverer.tools, cfg, self.pipeline.engine.device,
                      tables=tables, stats=stats)
            for _ in range(n)
        ]
        self._fe.set_native_span(
            [e._eng.span_handle() for e in self._span_engines],
            [c._cli.raw_handle() for c in self.pipeline.wire_clients],
            timeout_s=cfg.grpc.request_timeout_s,
            max_span_batch=cfg.gpu.max_batch,
            max_span_bytes=cfg.gpu.pinned_pool_bytes // 8,
            fallback_cb=self._fallback_cb,
        )
        log.info("native span enabled: %d engines, %d backends", n,
                 len(self.pipeline.wire_clients))

    def _fallback_cb(self, items) -> List[bytes]:
        """Rare slots the native span hands back: (kind, body, sid, headers,
        aux, tool_idx) per item — see span_api.h kinds.  Streaming and
        encode-side fallbacks re-enter the regular pipeline (nothing was
        invoked for them); decode-side fallbacks transcode the DELIVERED
        wire bytes in aux and never re-invoke."""
        from ..engine.batch import GpuPipeline

        out: List[bytes] = []
        timeout = self.config.grpc.request_timeout_s
        for kind, body, sid, hdr, aux, tool_idx in items:
            try:
                if kind == -1:
                    out.append(self._session_error(body, "session is blocked"))
                elif kind == -2:
                    out.append(self._session_error(
                        body, "session rate limit exceeded"))
                elif kind == 2:  # K_PY_NOT_TOOLCALL
                    out.append(self._handle_non_toolcall(body, hdr))
                elif kind == 5:  # K_PY_DEC_FALLBACK: delivered wire in aux
                    rid, _ = GpuPipeline._extract_id(body)
                    cpu_decode = getattr(self.pipeline, "_cpu_decode", None)
                    if cpu_decode is None:
                        raise RuntimeError("no decode fallback available")
                    st = self.pipeline.engine.stats
                    if hasattr(st, "host_fallbacks"):
                        st.host_fallbacks += 1
                    out.append(cpu_decode({"tool_idx": tool_idx}, bytes(aux),
                                          rid))
                elif kind in (3, 4):  # stream / encode fallback: not yet
                    # invoked — run the slot through the regular pipeline
                    out.append(self.pipeline.process_batch(
                        [body], headers=[hdr], timeout_s=timeout)[0])
                else:
                    out.append(self._session_error(body, "internal error"))
            except Exception as e:  # noqa: BLE001 - per-slot isolation
                from ..mcp import types as _mcp

                try:
                    rid, _ = GpuPipeline._extract_id(body)
                except Exception:
                    rid = None
                resp = _mcp.JSONRPCResponse(
                    id=rid,
                    error=_mcp.RPCError(_mcp.INTERNAL_ERROR, str(e)[:256]),
                )
                out.append(json.dumps(resp.to_dict()).encode())
        return out

    def start(self) -> int:
        self.port = self._fe.start()
        return self.port

    def stop(self, drain_s: float = 0.0) -> None:
        """drain_s > 0: graceful shutdown — stop accepting, finish
        in-flight requests (bounded), then stop (main.go:94-112 parity)."""
        if drain_s > 0 and hasattr(self._fe, "drain"):
            self._fe.drain(drain_s)
        self._fe.stop()
        if self._shard_pool is not None:
            self._shard_pool.shutdown(wait=False)

    # ---- hot path: one call per collected batch -----------------------------

    def _batch_cb(self, bodies: List[bytes], session_ids: List[Optional[str]],
                  headers: List[Dict[str, str]],
                  verdicts: Optional[List[int]] = None) -> List[Tuple[bytes, str]]:
        n = len(bodies)
        rejected: Dict[int, bytes] = {}
        if verdicts is not None:
            # C++ reactor already ran the session guard (session_table.h):
            # session_ids are final and verdicts carry blocked/rate-limit
            # decisions — the common all-allowed batch does ZERO per-request
            # Python session work here
            sids: List[str] = session_ids  # type: ignore[assignment]
            if self._cxx_header_filter:
                fwd_headers: List[Optional[Dict[str, str]]] = headers
            else:
                fwd_headers = [self.headers.filter_headers(h) for h in headers]
            for i, v in enumerate(verdicts):
                if v == 1:
                    rejected[i] = self._session_error(bodies[i], "session is blocked")
                elif v == 2:
                    rejected[i] = self._session_error(
                        bodies[i], "session rate limit exceeded")
            if rejected and not self._cxx_header_filter:
                fwd_headers = [h for i, h in enumerate(fwd_headers)
                               if i not in rejected]
            elif rejected:
                fwd_headers = [headers[i] for i in range(n) if i not in rejected]
        else:
            sids = []
            fwd_headers = []
            rl_on = self.config.session.rate_limit_enabled
            for i in range(n):
                # per-session guards (handler.go:219-226 parity) in one
                # manager-lock cycle (see SessionManager.guard)
                sess, verdict = self.sessions.guard(session_ids[i], headers[i],
                                                    rate_limit=rl_on)
                sids.append(sess.id)
                if verdict == 1:
                    rejected[i] = self._session_error(bodies[i], "session is blocked")
                    continue
                if verdict == 2:
                    rejected[i] = self._session_error(bodies[i], "session rate limit exceeded")
                    continue
                fwd_headers.append(headers[i] if self._cxx_header_filter
                                   else self.headers.filter_headers(headers[i]))
        timeout = self.config.grpc.request_timeout_s
        if rejected:
            live_idx = [i for i in range(n) if i not in rejected]
            live = [bodies[i] for i in live_idx]
            out_live = self._run_sharded(live, fwd_headers, sids, live_idx, timeout) if live else []
            out: List[bytes] = [b""] * n
            for k, i in enumerate(live_idx):
                out[i] = out_live[k]
            for i, resp in rejected.items():
                out[i] = resp
        else:
            out = self._run_sharded(bodies, fwd_headers, sids, None, timeout)
        return list(zip(out, sids))

    def _run_sharded(self, bodies, fwd_headers, sids, idx_map, timeout):
        """Split the batch across session shards (one pipeline per GPU)."""
        if self._shard_pool is None:
            return self.pipelines[0].process_batch(
                bodies, headers=fwd_headers, timeout_s=timeout
            )
        from ..parallel.dist import shard_for_session

        n_shards = len(self.pipelines)
        groups: List[List[int]] = [[] for _ in range(n_shards)]
        for k in range(len(bodies)):
            i = idx_map[k] if idx_map is not None else k
            groups[shard_for_session(sids[i], n_shards)].append(k)
        futs = []
        for s, g in enumerate(groups):
            if not g:
                futs.append(None)
                continue
            sub_bodies = [bodies[k] for k in g]
            sub_headers = [fwd_headers[k] for k in g] if fwd_headers else None
            futs.append(self._shard_pool.submit(
                self.pipelines[s].process_batch, sub_bodies,
                headers=sub_headers, timeout_s=timeout))
        out: List[bytes] = [b""] * len(bodies)
        for s, g in enumerate(groups):
            if futs[s] is None:
                continue
            sub_out = futs[s].result()
            for j, k in enumerate(g):
                out[k] = sub_out[j]
        return out

    @staticmethod
    def _session_error(body: bytes, message: str) -> bytes:
        try:
            rid = json.loads(body).get("id")
        except Exception:
            rid = None
        resp = mcp.JSONRPCResponse(
            id=rid, error=mcp.RPCError(mcp.INVALID_REQUEST, message)
        )
        return json.dumps(resp.to_dict()).encode()

    # ---- non-tools/call JSON-RPC on the batch path --------------------------

    def _handle_non_toolcall(self, body: bytes, hdr) -> bytes:
        try:
            data = json.loads(body)
        except Exception:
            resp = mcp.JSONRPCResponse(
                id=None, error=mcp.RPCError(mcp.PARSE_ERROR, "parse error")
            )
            return json.dumps(resp.to_dict()).encode()
        rid = data.get("id")
        method = data.get("method", "")
        if method == "initialize":
            result = self._initialize_result()
        elif method == "tools/list":
            # schema cache keyed by the tool-map version (the reference
            # declares a cache but never uses it, builder.go:18,29)
            self.tools.set_cache_key(getattr(self.discoverer, "tools_version", 0))
            result = {"tools": [t.to_dict() for t in
                                self.tools.build_tools(self.discoverer.get_methods())]}
        elif method == "prompts/list":
            result = {"prompts": []}
        elif method == "resources/list":
            result = {"resources": []}
        elif method == "notifications/initialized":
            result = {}
        else:
            resp = mcp.JSONRPCResponse(
                id=rid, error=mcp.RPCError(mcp.METHOD_NOT_FOUND,
                                           f"method not found: {method}")
            )
            return json.dumps(resp.to_dict()).encode()
        resp = mcp.JSONRPCResponse(id=rid, result=result)
        return json.dumps(resp.to_dict(), ensure_ascii=False).encode()

    def _initialize_result(self) -> Dict:
        return mcp.initialization_result()

    # ---- slow path: GET /, /health, /metrics, OPTIONS -----------------------

    def _slow_cb(self, method: str, path: str, headers: Dict[str, str],
                 body: bytes) -> Tuple[int, bytes, str]:
        if method == "OPTIONS":
            return 200, b"", ""
        if path == "/health":
            healthy = self.discoverer.health_check()
            count = len(self.discoverer.tools)
            ok = healthy and count > 0
            payload = json.dumps({
                "status": "healthy" if ok else "unhealthy",
                "timestamp": time.time(),
                "methodCount": count,
            }).encode()
            return (200 if ok else 503), payload, ""
        if path == "/metrics":
            stats = self.discoverer.stats()
            stats["sessions"] = self.sessions.stats()
            stats["uptimeS"] = time.time() - self.start_time
            stats["engine"] = self.pipeline.engine.stats.snapshot()
            if self._span_engines:
                # per-stage timers of the GIL-free serving path
                stats["nativeSpan"] = self._fe.native_stats()
            if len(self.pipelines) > 1:
                stats["shards"] = [p.engine.stats.snapshot() for p in self.pipelines]
            return 200, json.dumps(stats).encode(), ""
        if method == "GET" and path == "/":
            sess = self.sessions.get_or_create(
                headers.get("mcp-session-id"), headers
            )
            resp = mcp.JSONRPCResponse(id=None, result=self._initialize_result())
            return 200, json.dumps(resp.to_dict()).encode(), sess.id
        return 404, b'{"error":"not found"}', ""
