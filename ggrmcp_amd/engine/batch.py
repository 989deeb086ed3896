"""Batch pipeline driver: GPU engine + gRPC I/O + host fallback.

Two integration points replace the reference's per-request CPU hot path:

* ``GpuPipeline.process_batch(bodies)`` — the full GPU path used by the raw
  ingestion front end and bench.py: a batch of JSON-RPC request bodies goes
  through k_json2pb (envelope parse + validate + tool resolve + JSON->pb),
  the host fans the wire bytes out to the gRPC backend(s) concurrently
  (conn.Invoke stays host-side, reflection.go:367-373), and k_pb2json turns
  the response wire bytes into complete JSON-RPC response envelopes.

* ``BatchEngineInvoker`` — the async invoker seam of the HTTP handler
  (server/handler.py): concurrent tools/call coroutines are collected for up
  to ``batch_window_us`` and executed as value-mode GPU transcode batches.

Per-slot statuses from the kernels map to JSON-RPC errors; E_UNSUPPORTED /
E_OVERFLOW slots are re-transcoded on the CPU oracle (cpu_ref.py) and
counted — the GPU path never silently falls back wholesale.
"""

from __future__ import annotations

import asyncio
import json
import logging
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence, Tuple

import numpy as np

from ..config import Config
from ..mcp import types as mcp
from .cpu_ref import CpuTranscoder
from .tables import CompiledTables, compile_tables

log = logging.getLogger("ggrmcp.engine")

# status codes (mirror ops/csrc/common.h)
E_OK = 0
E_PARSE = 1
E_INVALID_REQUEST = 2
E_METHOD_NOT_FOUND = 3
E_INVALID_PARAMS = 4
E_LIMIT = 5
E_UNSUPPORTED = 6
E_OVERFLOW = 7
E_NOT_TOOLCALL = 8

SR_ID_IS_MISSING = 1
SR_SERVER_STREAMING = 2

_STATUS_TO_RPC = {
    E_PARSE: (mcp.PARSE_ERROR, "parse error"),
    E_INVALID_REQUEST: (mcp.INVALID_REQUEST, "invalid request"),
    E_METHOD_NOT_FOUND: (mcp.METHOD_NOT_FOUND, "tool not found"),
    E_INVALID_PARAMS: (mcp.INVALID_PARAMS, "invalid params"),
    E_LIMIT: (mcp.INVALID_PARAMS, "argument limits exceeded"),
    E_NOT_TOOLCALL: (mcp.METHOD_NOT_FOUND, "method not supported on batch path"),
}

SLOT_DTYPE = np.dtype(
    [
        ("status", "<i4"),
        ("tool_idx", "<i4"),
        ("pb_off", "<u4"),
        ("pb_len", "<u4"),
        ("err_pos", "<u4"),
        ("aux", "<i4"),
        ("id_len", "<u4"),
        ("flags", "<u4"),
    ]
)
DECODE_DTYPE = np.dtype(
    [("status", "<i4"), ("out_off", "<u4"), ("out_len", "<u4"), ("pad", "<u4")]
)


def _offsets(lengths: Sequence[int], pad: int = 0, align: int = 1) -> np.ndarray:
    caps = np.asarray(lengths, dtype=np.uint64) + pad
    if align > 1:
        caps = (caps + (align - 1)) // align * align
    off = np.zeros(len(caps) + 1, dtype=np.uint64)
    np.cumsum(caps, out=off[1:])
    return off.astype(np.uint32)


@dataclass
class EngineStats:
    batches: int = 0
    requests: int = 0
    gpu_ok: int = 0
    errors: int = 0
    host_fallbacks: int = 0
    encode_ns: int = 0
    decode_ns: int = 0
    invoke_ns: int = 0
    # device-only portions (copies+kernel+sync) of encode/decode — the
    # difference vs encode_ns/decode_ns is GIL/thread contention
    encode_gpu_ns: int = 0
    decode_gpu_ns: int = 0

    def snapshot(self) -> Dict[str, Any]:
        return {
            "batches": self.batches,
            "requests": self.requests,
            "gpuOk": self.gpu_ok,
            "errors": self.errors,
            "hostFallbacks": self.host_fallbacks,
            "encodeMs": self.encode_ns / 1e6,
            "decodeMs": self.decode_ns / 1e6,
            "invokeMs": self.invoke_ns / 1e6,
            "encodeGpuMs": self.encode_gpu_ns / 1e6,
            "decodeGpuMs": self.decode_gpu_ns / 1e6,
        }


class _SlotOfBatch:
    """Future-like view of one slot of a batched stream invoke."""

    def __init__(self, fut, index: int) -> None:
        self._fut = fut
        self._index = index

    def result(self):
        slots, res = self._fut.result()
        r = res[self._index]
        if isinstance(r, Exception):
            raise r
        return r


class GpuEngine:
    """One GPU device engine: compiled tables + the HIP extension Engine."""

    def __init__(self, tools: Dict[str, Any], config: Optional[Config] = None,
                 device: int = 0, tables: Optional[CompiledTables] = None,
                 stats: Optional["EngineStats"] = None) -> None:
        self.config = config or Config.default()
        from .. import ops

        self._mod = ops.load()
        if self._mod.device_count() == 0:
            raise ops.ExtensionUnavailable("no HIP device present")
        self.device = device
        self.tables: CompiledTables = tables if tables is not None else compile_tables(tools)
        self.tools = tools
        gpu = self.config.gpu
        # tool metadata for the fully-native span executor
        mis = [tools[name] for name in self.tables.tool_order]
        tool_paths = [mi.full_method_path for mi in mis]
        tool_out_msg = np.asarray(
            [self.tables.msg_index[mi.output_descriptor.full_name] for mi in mis],
            dtype=np.int32,
        )
        tool_backend = np.asarray([mi.backend_index for mi in mis], dtype=np.int32)
        self._eng = self._mod.Engine(
            device,
            self.tables.msg_table,
            self.tables.field_table,
            self.tables.enum_table,
            self.tables.enum_values,
            self.tables.tool_table,
            self.tables.name_blob,
            self.tables.n_msgs,
            self.tables.n_tools,
            max_batch=gpu.max_batch,
            cap_in=gpu.pinned_pool_bytes // 4,
            cap_pb=gpu.pinned_pool_bytes // 4,
            cap_scratch=gpu.device_pool_bytes // 4,
            cap_final=gpu.device_pool_bytes // 4,
            tool_paths=tool_paths,
            tool_out_msg=tool_out_msg,
            tool_backend=tool_backend,
        )
        self.stats = stats if stats is not None else EngineStats()
        # re-entrant: a span holds the engine for its whole encode ->
        # invoke -> decode sequence (mode-0 decode consumes the id slots of
        # THIS engine's last encode; interleaving another span's encode
        # between them would silently mis-id responses), while the inner
        # batch ops also take the lock for standalone callers
        self._lock = threading.RLock()

    # -- low-level batch ops -------------------------------------------------

    def encode_batch(
        self,
        payloads: Sequence[bytes],
        mode: int,
        msg_indices: Optional[Sequence[int]] = None,
        enforce: bool = True,
    ) -> Tuple[np.ndarray, List[Optional[bytes]]]:
        """Run k_json2pb. Returns (slot results, per-request pb bytes)."""
        msg_idx = (
            np.asarray(msg_indices, dtype=np.int32) if msg_indices is not None else None
        )
        t0 = time.perf_counter_ns()
        raw, pb_view = self._eng.encode_list(
            list(payloads),
            msg_idx,
            mode,
            10,
            1024,
            1 << 20,
            1 if enforce else 0,
        )
        self.stats.encode_ns += time.perf_counter_ns() - t0
        results = np.frombuffer(raw.tobytes(), dtype=SLOT_DTYPE)
        pb_mem = memoryview(pb_view)
        out: List[Optional[bytes]] = []
        for r in results:
            if r["status"] == E_OK:
                out.append(bytes(pb_mem[r["pb_off"] : r["pb_off"] + r["pb_len"]]))
            else:
                out.append(None)
        return results, out

    def decode_batch(
        self,
        payloads: Sequence[Optional[bytes]],
        msg_indices: Sequence[int],
        mode: int,
        skip: Optional[Sequence[bool]] = None,
    ) -> Tuple[np.ndarray, List[Optional[bytes]]]:
        """Run k_pb2json. ``None`` payloads are auto-skipped slots."""
        skips = [1 if (p is None or (skip is not None and skip[i])) else 0
                 for i, p in enumerate(payloads)]
        t0 = time.perf_counter_ns()
        raw, out_view = self._eng.decode_list(
            list(payloads),
            np.asarray(msg_indices, dtype=np.int32),
            np.asarray(skips, dtype=np.int32),
            mode,
        )
        self.stats.decode_ns += time.perf_counter_ns() - t0
        results = np.frombuffer(raw.tobytes(), dtype=DECODE_DTYPE)
        mem = memoryview(out_view)
        out: List[Optional[bytes]] = []
        for i, r in enumerate(results):
            if skips[i] or r["status"] != E_OK:
                out.append(None)
            else:
                out.append(bytes(mem[r["out_off"] : r["out_off"] + r["out_len"]]))
        return results, out


class GpuPipeline:
    """Full tools/call pipeline over one GpuEngine + a ServiceDiscoverer."""

    def __init__(self, discoverer, config: Optional[Config] = None, device: int = 0,
                 invoke_workers: int = 64, wire_clients=None) -> None:
        self.discoverer = discoverer
        self.config = config or Config.default()
        # config.gpu.streams engine instances (each its own HIP stream +
        # arenas): concurrent chunks overlap copies/kernels/invokes, hiding
        # the host I/O stage behind the GPU stages and vice versa
        n_engines = max(1, self.config.gpu.streams)
        shared_stats = EngineStats()
        tables = compile_tables(discoverer.tools)
        self.engines = [
            GpuEngine(discoverer.tools, self.config, device, tables=tables,
                      stats=shared_stats)
            for _ in range(n_engines)
        ]
        self.engine = self.engines[0]
        self.cpu = CpuTranscoder()
        # native C++ h2 transport per backend index (None -> grpcio threads)
        self.wire_clients = wire_clients
        self._invoke_pool = ThreadPoolExecutor(
            max_workers=invoke_workers, thread_name_prefix="ginvoke"
        )
        self._chunk_pool = ThreadPoolExecutor(
            max_workers=max(2 * n_engines, 4), thread_name_prefix="gchunk"
        )
        self._engine_rr = 0
        # optional: handles well-formed JSON-RPC envelopes whose method is
        # NOT tools/call (initialize, tools/list, ...) on the batch path;
        # (body: bytes, headers: dict|None) -> response bytes
        self.non_toolcall_handler = None
        # tool idx -> MethodInfo
        self._mi_by_idx = [
            discoverer.tools[name] for name in self.engine.tables.tool_order
        ]
        self._out_msg_idx = np.asarray(
            [
                self.engine.tables.msg_index[mi.output_descriptor.full_name]
                for mi in self._mi_by_idx
            ],
            dtype=np.int32,
        )

    def close(self) -> None:
        self._invoke_pool.shutdown(wait=False)
        self._chunk_pool.shutdown(wait=False)

    # ---- the full batched hot path ----------------------------------------

    def process_batch(
        self,
        bodies: Sequence[bytes],
        headers: Optional[Sequence[Dict[str, str]]] = None,
        timeout_s: Optional[float] = None,
    ) -> List[bytes]:
        """JSON-RPC request bodies -> JSON-RPC response bodies.

        Large batches split across the engine instances; chunks run
        concurrently so one chunk's gRPC invoke overlaps another's GPU
        encode/decode (copy/compute/IO pipelining across HIP streams)."""
        n = len(bodies)
        n_eng = len(self.engines)
        min_chunk = 64
        if n_eng == 1 or n < 2 * min_chunk:
            return self._process_span(self.engine, bodies, headers, timeout_s)
        n_chunks = min(n_eng, (n + min_chunk - 1) // min_chunk)
        bounds = [round(i * n / n_chunks) for i in range(n_chunks + 1)]
        # rotate the starting engine so concurrent batches (two frontend
        # workers) land on disjoint engines instead of all serializing on
        # engines[0]
        base = self._engine_rr
        self._engine_rr = (base + n_chunks) % n_eng
        futs = []
        for k in range(n_chunks):
            lo, hi = bounds[k], bounds[k + 1]
            hdr = headers[lo:hi] if headers else None
            futs.append(
                self._chunk_pool.submit(
                    self._process_span, self.engines[(base + k) % n_eng],
                    bodies[lo:hi], hdr, timeout_s
                )
            )
        out: List[bytes] = []
        for f in futs:
            out.extend(f.result())
        return out

    def _process_span(
        self,
        engine: GpuEngine,
        bodies: Sequence[bytes],
        headers: Optional[Sequence[Dict[str, str]]] = None,
        timeout_s: Optional[float] = None,
    ) -> List[bytes]:
        with engine._lock:
            return self._process_span_locked(engine, bodies, headers, timeout_s)

    def _process_span_locked(
        self,
        engine: GpuEngine,
        bodies: Sequence[bytes],
        headers: Optional[Sequence[Dict[str, str]]] = None,
        timeout_s: Optional[float] = None,
    ) -> List[bytes]:
        st = engine.stats
        st.batches += 1
        st.requests += len(bodies)
        if self.wire_clients:
            return self._native_span(engine, bodies, headers, timeout_s)
        enc, pbs = engine.encode_batch(bodies, mode=0)

        # fan out gRPC invocations for OK slots (host-side I/O stage)
        n = len(bodies)
        resp_wire: List[Optional[bytes]] = [None] * n
        rpc_error: List[Optional[Exception]] = [None] * n
        out_idx = np.zeros(n, dtype=np.int32)

        # server-streaming slots: fan out unary->stream invokes concurrently;
        # their wire chunks decode in one extra GPU batch below (config 4).
        # (The native-transport batching variant lives in _native_span; this
        # path always runs on grpcio threads — _process_span_locked returns
        # early when wire_clients is set.)
        stream_futs: Dict[int, Any] = {}
        timeout0 = timeout_s if timeout_s is not None else self.config.grpc.request_timeout_s
        for i in range(n):
            if enc[i]["status"] == E_OK and enc[i]["flags"] & SR_SERVER_STREAMING:
                mi = self._mi_by_idx[enc[i]["tool_idx"]]
                hdr = headers[i] if headers else None
                stream_futs[i] = self._invoke_pool.submit(
                    self.discoverer.invoke_streaming_wire, mi, pbs[i], hdr, timeout0
                )

        t0 = time.perf_counter_ns()
        futures = {}
        for i in range(n):
            if enc[i]["status"] != E_OK:
                continue
            if enc[i]["flags"] & SR_SERVER_STREAMING:
                continue  # streaming handled below via host assembly
            mi = self._mi_by_idx[enc[i]["tool_idx"]]
            out_idx[i] = self._out_msg_idx[enc[i]["tool_idx"]]
            hdr = headers[i] if headers else None
            futures[i] = self._invoke_pool.submit(
                self.discoverer.invoke_wire, mi, pbs[i], hdr, timeout_s
            )
        for i, fut in futures.items():
            try:
                resp_wire[i] = fut.result()
            except Exception as e:
                rpc_error[i] = e
        st.invoke_ns += time.perf_counter_ns() - t0

        dec, finals = engine.decode_batch(resp_wire, out_idx, mode=0)

        # streaming: gather chunk lists, decode ALL chunks of ALL streams in
        # one value-mode GPU batch, assemble envelopes host-side
        stream_out: Dict[int, bytes] = {}
        if stream_futs:
            stream_out = self._decode_streams(engine, stream_futs, enc, bodies)

        # assemble the batch: GPU envelopes where OK, host for the rest
        out: List[bytes] = []
        for i in range(n):
            if finals[i] is not None:
                st.gpu_ok += 1
                out.append(finals[i])
                continue
            if i in stream_out:
                out.append(stream_out[i])
                continue
            out.append(self._host_slot(engine, bodies[i], enc[i], dec[i] if resp_wire[i] is not None else None,
                                       resp_wire[i], rpc_error[i], headers[i] if headers else None,
                                       timeout_s))
        return out

    def _decode_streams(self, engine, stream_futs, enc, bodies) -> Dict[int, bytes]:
        """Batch-decode every stream chunk on the GPU (mode 1) and wrap each
        stream's chunks as the ToolCallResult content list (the capability
        the reference rejects outright, discovery.go:354-356)."""
        st = engine.stats
        chunks_by_slot: Dict[int, List[bytes]] = {}
        errors: Dict[int, Exception] = {}
        for i, fut in stream_futs.items():
            try:
                chunks_by_slot[i] = fut.result()
            except Exception as e:
                errors[i] = e
        flat: List[bytes] = []
        flat_idx: List[int] = []
        spans: Dict[int, Tuple[int, int]] = {}
        for i, chunks in chunks_by_slot.items():
            tool = int(enc[i]["tool_idx"])
            spans[i] = (len(flat), len(chunks))
            flat.extend(chunks)
            flat_idx.extend([int(self._out_msg_idx[tool])] * len(chunks))
        jsons: List[Optional[bytes]] = []
        cap = engine._eng.max_batch
        for base in range(0, len(flat), cap):
            with engine._lock:
                _, part = engine.decode_batch(
                    flat[base : base + cap], flat_idx[base : base + cap], mode=2
                )
            jsons.extend(part)
        out: Dict[int, bytes] = {}
        for i in stream_futs:
            rid, _ = self._extract_id(bodies[i])
            if i in errors:
                e = errors[i]
                if hasattr(e, "code") and callable(e.code):
                    text = f"gRPC error {e.code().name}: {e.details() if hasattr(e, 'details') else e}"
                else:
                    text = str(e)
                st.errors += 1
                result = mcp.ToolCallResult(content=[mcp.TextContent(text)], is_error=True)
                resp = mcp.JSONRPCResponse(id=rid, result=result.to_dict())
                out[i] = json.dumps(resp.to_dict(), ensure_ascii=False).encode()
                continue
            start, count = spans[i]
            items: List[bytes] = []
            ok = True
            mi = self._mi_by_idx[int(enc[i]["tool_idx"])]
            for k in range(start, start + count):
                if jsons[k] is not None:
                    # already a complete escaped {"type":"text","text":"..."}
                    items.append(jsons[k])
                else:
                    # per-chunk CPU fallback (counted, never silent)
                    st.host_fallbacks += 1
                    try:
                        t = self.cpu.pb_to_json(mi.output_descriptor, flat[k])
                        items.append(
                            b'{"type":"text","text":'
                            + json.dumps(t, ensure_ascii=False).encode()
                            + b"}"
                        )
                    except Exception as e:
                        st.errors += 1
                        resp = mcp.JSONRPCResponse(
                            id=rid, error=mcp.RPCError(mcp.INTERNAL_ERROR, str(e)[:512])
                        )
                        out[i] = json.dumps(resp.to_dict(), ensure_ascii=False).encode()
                        ok = False
                        break
            if not ok:
                continue
            st.gpu_ok += 1
            out[i] = (
                b'{"jsonrpc":"2.0","id":'
                + json.dumps(rid).encode()
                + b',"result":{"content":['
                + b",".join(items)
                + b'],"isError":false}}'
            )
        return out


    def _native_span(
        self,
        engine: GpuEngine,
        bodies: Sequence[bytes],
        headers: Optional[Sequence[Dict[str, str]]],
        timeout_s: Optional[float],
    ) -> List[bytes]:
        """encode -> invoke -> decode entirely in C++ (engine.process_span);
        Python only assembles error/streaming/fallback slots."""
        st = engine.stats
        timeout = timeout_s if timeout_s is not None else self.config.grpc.request_timeout_s
        t0 = time.perf_counter_ns()
        (enc_raw, dec_raw, out_view, stream_pbs, rpc_errs,
         resp_wires) = engine._eng.process_span(
            list(bodies),
            list(headers) if headers else None,
            [c._cli for c in self.wire_clients],
            timeout,
        )
        total_ns = time.perf_counter_ns() - t0
        # per-stage split from the C++ span (SURVEY §5 /metrics deliverable);
        # Python call overhead stays booked under invoke so stages sum to total
        enc_ms, _inv_ms, dec_ms = engine._eng.last_stage_ms()
        st.encode_ns += int(enc_ms * 1e6)
        st.decode_ns += int(dec_ms * 1e6)
        st.invoke_ns += max(0, total_ns - int(enc_ms * 1e6) - int(dec_ms * 1e6))
        enc_gpu_ms, dec_gpu_ms = engine._eng.last_gpu_ms()
        st.encode_gpu_ns += int(enc_gpu_ms * 1e6)
        st.decode_gpu_ns += int(dec_gpu_ms * 1e6)
        enc = np.frombuffer(enc_raw.tobytes(), dtype=SLOT_DTYPE)
        dec = np.frombuffer(dec_raw.tobytes(), dtype=DECODE_DTYPE)
        n = len(bodies)
        mem = memoryview(out_view)
        # materialize unary responses NOW: the stream-chunk decode below
        # reuses the engine's pinned output buffer and would overwrite them
        finals: List[Optional[bytes]] = [None] * n
        for i in range(n):
            r = dec[i]
            if (r["status"] == E_OK and r["out_len"] > 0 and rpc_errs[i] is None
                    and enc[i]["status"] == E_OK
                    and not (enc[i]["flags"] & SR_SERVER_STREAMING)):
                finals[i] = bytes(mem[r["out_off"] : r["out_off"] + r["out_len"]])

        # streaming slots: batched native stream invoke + GPU chunk decode
        stream_futs: Dict[int, Any] = {}
        stream_per_be: Dict[int, List[Any]] = {}
        for i in range(n):
            if enc[i]["status"] == E_OK and enc[i]["flags"] & SR_SERVER_STREAMING:
                mi = self._mi_by_idx[enc[i]["tool_idx"]]
                hdr = headers[i] if headers else None
                be = mi.backend_index if mi.backend_index < len(self.wire_clients) else 0
                g = stream_per_be.setdefault(be, [[], [], [], []])
                g[0].append(i)
                g[1].append(mi.full_method_path)
                g[2].append(stream_pbs[i])
                g[3].append(list(hdr.items()) if hdr else [])
        for be, g in stream_per_be.items():
            def run_stream_backend(be=be, g=g):
                return g[0], self.wire_clients[be].invoke_stream_batch(
                    g[1], g[2], timeout, g[3]
                )
            fut = self._invoke_pool.submit(run_stream_backend)
            for k, i in enumerate(g[0]):
                stream_futs[i] = _SlotOfBatch(fut, k)
        stream_out: Dict[int, bytes] = {}
        if stream_futs:
            stream_out = self._decode_streams(engine, stream_futs, enc, bodies)

        out: List[bytes] = []
        for i in range(n):
            if finals[i] is not None:
                st.gpu_ok += 1
                out.append(finals[i])
                continue
            if i in stream_out:
                out.append(stream_out[i])
                continue
            err = rpc_errs[i]
            rpc_exc = None
            wire = resp_wires[i]  # delivered bytes when GPU decode failed
            if err is not None:
                from ..backend.native_invoker import NativeRpcError

                rpc_exc = NativeRpcError(int(err[0]), err[1])
                wire = None
            out.append(
                self._host_slot(
                    engine, bodies[i], enc[i],
                    dec[i] if wire is not None else None,
                    wire, rpc_exc,
                    headers[i] if headers else None, timeout_s,
                )
            )
        return out

    # ---- host handling of non-GPU slots ------------------------------------

    def _host_slot(self, engine, body, enc_r, dec_r, wire, rpc_err, hdr, timeout_s) -> bytes:
        st = engine.stats
        rid, has_id = self._extract_id(body)
        status = int(enc_r["status"])
        flags = int(enc_r["flags"])

        if status == E_OK and flags & SR_SERVER_STREAMING:
            return self._host_streaming(body, enc_r, rid, hdr, timeout_s)

        if status == E_NOT_TOOLCALL and self.non_toolcall_handler is not None:
            try:
                return self.non_toolcall_handler(body, hdr)
            except Exception as e:
                st.errors += 1
                resp = mcp.JSONRPCResponse(
                    id=rid, error=mcp.RPCError(mcp.INTERNAL_ERROR, str(e)[:256])
                )
                return json.dumps(resp.to_dict(), ensure_ascii=False).encode()

        if status in (E_UNSUPPORTED, E_OVERFLOW):
            # encode-side fallback: nothing was invoked yet, so the CPU path
            # performs the one and only invoke (counted)
            st.host_fallbacks += 1
            return self._cpu_full(body, rid, hdr, timeout_s)

        if status == E_OK and rpc_err is None and wire is not None:
            # decode-side fallback: the response WAS delivered; transcode the
            # received bytes on the CPU oracle — never re-invoke, which would
            # duplicate side effects on non-idempotent methods
            st.host_fallbacks += 1
            return self._cpu_decode(enc_r, bytes(wire), rid)

        if status == E_OK and rpc_err is not None:
            # gRPC failure -> isError tool result (handler.go:252-259);
            # NativeRpcError duck-types grpc.RpcError (code().name/details())
            if hasattr(rpc_err, "code") and callable(rpc_err.code):
                code = rpc_err.code().name
                detail = rpc_err.details() if hasattr(rpc_err, "details") else str(rpc_err)
                text = f"gRPC error {code}: {detail}"
            else:
                text = str(rpc_err)
            st.errors += 1
            result = mcp.ToolCallResult(content=[mcp.TextContent(text)], is_error=True)
            resp = mcp.JSONRPCResponse(id=rid, result=result.to_dict())
            return json.dumps(resp.to_dict(), ensure_ascii=False).encode()

        code, msg = _STATUS_TO_RPC.get(status, (mcp.INTERNAL_ERROR, "internal error"))
        st.errors += 1
        resp = mcp.JSONRPCResponse(id=rid, error=mcp.RPCError(code, msg))
        return json.dumps(resp.to_dict(), ensure_ascii=False).encode()

    def _cpu_decode(self, enc_r, wire: bytes, rid) -> bytes:
        """Transcode an already-received response wire on the CPU (protojson
        semantics via cpu_ref) and wrap it as the tool-call result envelope.
        Used when the GPU decode stage rejects a delivered response (invalid
        UTF-8, over-cap, E_PARSE/E_LIMIT) — the RPC is NOT repeated."""
        mi = self._mi_by_idx[int(enc_r["tool_idx"])]
        try:
            text = self.cpu.pb_to_json(mi.output_descriptor, wire)
            result = mcp.ToolCallResult(
                content=[mcp.TextContent(text)], is_error=False
            )
            resp = mcp.JSONRPCResponse(id=rid, result=result.to_dict())
        except Exception as e:
            resp = mcp.JSONRPCResponse(
                id=rid, error=mcp.RPCError(mcp.INTERNAL_ERROR, str(e)[:512])
            )
        return json.dumps(resp.to_dict(), ensure_ascii=False).encode()

    def _cpu_full(self, body: bytes, rid, hdr, timeout_s) -> bytes:
        try:
            data = json.loads(body)
            params = data.get("params") or {}
            tool = params.get("name", "")
            args_json = json.dumps(params.get("arguments", {}), ensure_ascii=False)
            output = self.discoverer.invoke_method_by_tool(tool, args_json, hdr, timeout_s)
            result = mcp.ToolCallResult(content=[mcp.TextContent(output)], is_error=False)
            resp = mcp.JSONRPCResponse(id=rid, result=result.to_dict())
        except Exception as e:
            resp = mcp.JSONRPCResponse(
                id=rid, error=mcp.RPCError(mcp.INTERNAL_ERROR, str(e)[:512])
            )
        return json.dumps(resp.to_dict(), ensure_ascii=False).encode()

    def _host_streaming(self, body, enc_r, rid, hdr, timeout_s) -> bytes:
        mi = self._mi_by_idx[enc_r["tool_idx"]]
        try:
            data = json.loads(body)
            args_json = json.dumps(
                (data.get("params") or {}).get("arguments", {}), ensure_ascii=False
            )
            chunks = list(
                self.discoverer.invoke_streaming(mi.tool_name(), args_json, hdr, timeout_s)
            )
            result = mcp.ToolCallResult(
                content=[mcp.TextContent(c) for c in chunks], is_error=False
            )
            resp = mcp.JSONRPCResponse(id=rid, result=result.to_dict())
        except Exception as e:
            resp = mcp.JSONRPCResponse(
                id=rid, error=mcp.RPCError(mcp.INTERNAL_ERROR, str(e)[:512])
            )
        return json.dumps(resp.to_dict(), ensure_ascii=False).encode()

    @staticmethod
    def _extract_id(body: bytes):
        try:
            data = json.loads(body)
            return data.get("id"), "id" in data
        except Exception:
            return None, False


def build_wire_clients(discoverer, config: Config):
    """Native h2 batch clients for every backend (serving hot path).  Falls
    back to None (grpcio threads) when the extension is unavailable."""
    try:
        from ..backend.native_invoker import NativeWireClient
    except Exception:  # pragma: no cover
        return None
    clients = []
    n_conns = getattr(config.grpc, "native_connections", 8) or 8
    # reference connection.go:55-57's 4 MB recv cap, enforced per CALL so
    # one oversized backend response errors alone instead of blowing the
    # whole batch's response arena
    max_resp = getattr(config.grpc, "max_recv_msg_bytes", 0) or 0
    try:
        for conn in discoverer.connections:
            clients.append(NativeWireClient(conn.target, connections=n_conns,
                                            max_resp_bytes=max_resp))
    except Exception as e:  # pragma: no cover - fall back to grpcio
        log.warning("native transport unavailable (%s); using grpcio", e)
        for c in clients:
            c.close()
        return None
    return clients


class BatchEngineInvoker:
    """Async invoker seam for the HTTP handler: batches concurrent value-mode
    transcodes onto the GPU (the handler already parsed the envelope)."""

    def __init__(self, discoverer, config: Optional[Config] = None, device: int = 0,
                 wire_clients=None):
        self.config = config or Config.default()
        if wire_clients is None:
            wire_clients = build_wire_clients(discoverer, self.config)
        self.pipeline = GpuPipeline(discoverer, self.config, device,
                                    wire_clients=wire_clients)
        self.discoverer = discoverer
        self._queue: List[Tuple[str, str, Dict[str, str], float, asyncio.Future]] = []
        self._qlock = threading.Lock()
        self._wakeup: Optional[asyncio.Event] = None
        self._task: Optional[asyncio.Task] = None

    def stats(self) -> Dict[str, Any]:
        return self.pipeline.engine.stats.snapshot()

    async def _ensure_worker(self) -> None:
        if self._task is None or self._task.done():
            self._wakeup = asyncio.Event()
            self._task = asyncio.get_running_loop().create_task(self._worker())

    async def invoke(self, tool_name, args_json, headers, timeout_s) -> str:
        await self._ensure_worker()
        fut = asyncio.get_running_loop().create_future()
        with self._qlock:
            self._queue.append((tool_name, args_json, headers, timeout_s, fut))
        self._wakeup.set()
        return await fut

    async def invoke_streaming(self, tool_name, args_json, headers, timeout_s):
        loop = asyncio.get_running_loop()
        return await loop.run_in_executor(
            self.pipeline._invoke_pool,
            lambda: list(
                self.discoverer.invoke_streaming(tool_name, args_json, headers, timeout_s)
            ),
        )

    async def _worker(self) -> None:
        window_s = self.config.gpu.batch_window_us / 1e6
        while True:
            await self._wakeup.wait()
            await asyncio.sleep(window_s)  # batch collection window
            with self._qlock:
                batch = self._queue
                self._queue = []
                self._wakeup.clear()
            if not batch:
                continue
            loop = asyncio.get_running_loop()
            try:
                results = await loop.run_in_executor(None, self._run_batch, batch)
                for (_, _, _, _, fut), res in zip(batch, results):
                    if not fut.done():
                        if isinstance(res, Exception):
                            fut.set_exception(res)
                        else:
                            fut.set_result(res)
            except Exception as e:  # batch-level failure
                for _, _, _, _, fut in batch:
                    if not fut.done():
                        fut.set_exception(e)

    def _run_batch(self, batch) -> List[Any]:
        """Value-mode GPU transcode for a set of already-validated calls."""
        pipeline = self.pipeline
        eng = pipeline.engine
        eng.stats.batches += 1
        eng.stats.requests += len(batch)
        in_idx = np.zeros(len(batch), dtype=np.int32)
        mis = []
        payloads = []
        for i, (tool_name, args_json, _h, _t, _f) in enumerate(batch):
            mi = self.discoverer.tools.get(tool_name)
            mis.append(mi)
            if mi is None:
                payloads.append(b"{}")
                continue
            in_idx[i] = eng.tables.msg_index[mi.input_descriptor.full_name]
            payloads.append(args_json.encode())
        enc, pbs = eng.encode_batch(payloads, mode=1, msg_indices=in_idx, enforce=True)
        results: List[Any] = [None] * len(batch)
        out_idx = np.zeros(len(batch), dtype=np.int32)
        wires: List[Optional[bytes]] = [None] * len(batch)
        futures = {}
        ok_slots = set()
        per_be: Dict[int, List[Any]] = {}
        wcs = pipeline.wire_clients
        t0 = time.perf_counter_ns()
        for i, (tool_name, args_json, hdr, timeout_s, _f) in enumerate(batch):
            mi = mis[i]
            if mi is None:
                from ..backend.discovery import MethodNotFoundError

                results[i] = MethodNotFoundError(f"tool not found: {tool_name}")
                continue
            if enc[i]["status"] == E_OK:
                out_idx[i] = eng.tables.msg_index[mi.output_descriptor.full_name]
                ok_slots.add(i)
                if wcs:
                    be = mi.backend_index if mi.backend_index < len(wcs) else 0
                    g = per_be.setdefault(be, [[], [], [], [], []])
                    g[0].append(i)
                    g[1].append(mi.full_method_path)
                    g[2].append(pbs[i])
                    g[3].append(list(hdr.items()) if hdr else [])
                    g[4].append(timeout_s)
                else:
                    futures[i] = pipeline._invoke_pool.submit(
                        self.discoverer.invoke_wire, mi, pbs[i], hdr, timeout_s
                    )
            elif enc[i]["status"] in (E_UNSUPPORTED, E_OVERFLOW):
                eng.stats.host_fallbacks += 1
                try:
                    results[i] = self.discoverer.invoke_method_by_tool(
                        tool_name, args_json, hdr, timeout_s
                    )
                except Exception as e:
                    results[i] = e
            else:
                from ..mcp.validation import ValidationError

                code, msg = _STATUS_TO_RPC.get(
                    int(enc[i]["status"]), (mcp.INVALID_PARAMS, "invalid arguments")
                )
                eng.stats.errors += 1
                results[i] = ValidationError("arguments", msg)
        if per_be:
            def run_backend(be, g):
                return g[0], wcs[be].invoke_batch(g[1], g[2], max(g[4]), g[3])

            batch_futs = [
                pipeline._invoke_pool.submit(run_backend, be, g)
                for be, g in per_be.items()
            ]
            for fut in batch_futs:
                slots, res = fut.result()
                for i, r in zip(slots, res):
                    if isinstance(r, Exception):
                        results[i] = r
                    else:
                        wires[i] = r
        for i, fut in futures.items():
            try:
                wires[i] = fut.result()
            except Exception as e:
                results[i] = e
        eng.stats.invoke_ns += time.perf_counter_ns() - t0
        dec, jsons = eng.decode_batch(wires, out_idx, mode=1)
        for i in ok_slots:
            if wires[i] is None:
                continue
            if jsons[i] is not None:
                eng.stats.gpu_ok += 1
                results[i] = jsons[i].decode()
            else:
                eng.stats.host_fallbacks += 1
                mi = mis[i]
                try:
                    results[i] = pipeline.cpu.pb_to_json(mi.output_descriptor, wires[i])
                except Exception as e:
                    results[i] = e
        return results
