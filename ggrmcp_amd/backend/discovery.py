"""Service discovery + dynamic invocation orchestrator.

Re-design of the reference's ``pkg/grpc/discovery.go``: owns the connection
layer and both discovery sources (descriptor-set-first with reflection
fallback, discovery.go:101-119), publishes the tool map atomically
(discovery.go:122-127 — a Python dict swap has the same lock-free-read
semantics as the reference's ``atomic.Pointer``), retries reconnects
(discovery.go:186-235), exposes stats (discovery.go:278-333), and implements
the per-call string seam ``invoke_method_by_tool`` (discovery.go:346-369 +
reflection.go:333-391: metadata from filtered headers -> dynamic message ->
protojson decode -> unary invoke -> protojson encode).

MI355X extensions over the reference:
* several backends (centralized-gateway mode) — each MethodInfo records its
  ``backend_index``;
* server-streaming invocation (the reference rejects it,
  discovery.go:354-356) via ``invoke_streaming``;
* ``descriptor_blob()`` — a deterministic serialized snapshot of the
  discovered descriptor state, broadcast to the other GPU ranks over RCCL so
  every shard compiles identical GPU transcode tables (parallel/dist.py).
"""

from __future__ import annotations

import logging
import threading
import time
from typing import Any, Dict, Iterator, List, Mapping, Optional, Sequence, Tuple

import grpc
from google.protobuf import descriptor_pb2, json_format, message_factory

from ..config import Config, ConnectionConfig
from ..descriptors.loader import DescriptorLoader, build_pool, extract_method_infos
from ..types import MethodInfo
from .connection import ConnectionManager
from .reflection import ReflectionClient

log = logging.getLogger("ggrmcp.discovery")

STREAMING_UNSUPPORTED_MSG = "streaming methods are not supported"  # ref discovery.go:354-356


class MethodNotFoundError(KeyError):
    pass


class ServiceDiscoverer:
    """Reference serviceDiscoverer (pkg/grpc/discovery.go)."""

    def __init__(
        self,
        config: Optional[Config] = None,
        backends: Optional[Sequence[ConnectionConfig]] = None,
    ) -> None:
        self.config = config or Config.default()
        backend_cfgs = list(backends) if backends is not None else self.config.all_backends()
        self.connections: List[ConnectionManager] = [
            ConnectionManager(c) for c in backend_cfgs
        ]
        self.reflection_clients: List[Optional[ReflectionClient]] = [None] * len(
            self.connections
        )
        # tools map published by atomic reference swap (discovery.go:122-127)
        self._tools: Dict[str, MethodInfo] = {}
        self._tools_version = 0
        # serialized FileDescriptorProtos per backend for descriptor_blob()
        self._fdps: List[List[descriptor_pb2.FileDescriptorProto]] = [
            [] for _ in self.connections
        ]
        self._msg_cls_cache: Dict[int, Any] = {}
        self._lock = threading.Lock()
        self._invocations = 0
        self._errors = 0

    # -- connection (discovery.go:65-89) -------------------------------------

    def connect(self, timeout_s: Optional[float] = None) -> None:
        for i, conn in enumerate(self.connections):
            channel = conn.connect(timeout_s)
            self.reflection_clients[i] = ReflectionClient(channel)

    def close(self) -> None:
        for conn in self.connections:
            conn.close()

    # -- discovery (discovery.go:91-168) --------------------------------------

    def discover(self) -> Dict[str, MethodInfo]:
        """Descriptor-set first (when enabled), reflection fallback
        (discovery.go:101-119); merge across backends; atomic publish."""
        all_infos: List[MethodInfo] = []
        ds = self.config.descriptor_set
        used_descriptor_set = False
        if ds.enabled and ds.path:
            try:
                loader = DescriptorLoader(ds.path).load()
                infos = loader.extract_method_info(backend_index=0)
                self._fdps[0] = list(loader.fdset.file)
                all_infos.extend(infos)
                used_descriptor_set = True
            except Exception as e:  # fall back to reflection (discovery.go:107-111)
                log.warning("descriptor set load failed (%s); falling back to reflection", e)
        for i, rc in enumerate(self.reflection_clients):
            if i == 0 and used_descriptor_set:
                continue
            if rc is None:
                continue
            infos = rc.discover_methods(backend_index=i)
            with rc._lock:
                self._fdps[i] = list(rc._fd_cache.values())
            all_infos.extend(infos)
        tools: Dict[str, MethodInfo] = {}
        for mi in all_infos:
            name = mi.tool_name()
            if name in tools:
                log.warning("duplicate tool name %s; keeping first", name)
                continue
            tools[name] = mi
        self.publish_tools(tools)
        log.info("discovered %d tools", len(tools))
        return tools

    def publish_tools(self, tools: Dict[str, MethodInfo]) -> None:
        """Atomic swap (discovery.go:122-127)."""
        self._tools = dict(tools)
        self._tools_version += 1

    # -- accessors -----------------------------------------------------------

    @property
    def tools(self) -> Dict[str, MethodInfo]:
        return self._tools  # reference-snapshot read; never mutated in place

    @property
    def tools_version(self) -> int:
        return self._tools_version

    def get_methods(self) -> List[MethodInfo]:
        return list(self._tools.values())

    def get_method_by_tool(self, tool_name: str) -> MethodInfo:
        mi = self._tools.get(tool_name)
        if mi is None:
            raise MethodNotFoundError(f"tool not found: {tool_name}")
        return mi

    _BLOB_MAGIC = b"GGDB"  # framed multi-backend snapshot

    def descriptor_blob(self) -> bytes:
        """Deterministic serialized descriptor snapshot (for RCCL broadcast).

        Framed per backend so centralized-gateway mode survives the
        broadcast: [magic][u32 n][u32 len_i][fdset_i]...  Each fdset keeps
        its backend's files in sorted order."""
        import struct

        parts = []
        for backend_fdps in self._fdps:
            fdset = descriptor_pb2.FileDescriptorSet()
            for fdp in sorted(backend_fdps, key=lambda f: f.name):
                fdset.file.append(fdp)
            parts.append(fdset.SerializeToString())
        out = [self._BLOB_MAGIC, struct.pack("<I", len(parts))]
        for p in parts:
            out.append(struct.pack("<I", len(p)))
            out.append(p)
        return b"".join(out)

    def load_descriptor_blob(
        self, blob: bytes, backend_index: int = 0, merge: bool = True
    ) -> None:
        """Rebuild the tool map from a snapshot (non-rank-0 shards, or the
        descriptor-set path for one backend).  Accepts a raw
        FileDescriptorSet (single backend, placed at ``backend_index``) or
        the framed multi-backend form produced by ``descriptor_blob``."""
        import struct

        per_backend: List[Tuple[int, bytes]] = []
        if blob.startswith(self._BLOB_MAGIC):
            off = len(self._BLOB_MAGIC)
            (count,) = struct.unpack_from("<I", blob, off)
            off += 4
            for b in range(count):
                (ln,) = struct.unpack_from("<I", blob, off)
                off += 4
                per_backend.append((b, blob[off : off + ln]))
                off += ln
        else:
            per_backend.append((backend_index, blob))

        tools: Dict[str, MethodInfo] = dict(self._tools) if merge else {}
        for b, raw in per_backend:
            fdset = descriptor_pb2.FileDescriptorSet.FromString(raw)
            if not fdset.file:
                continue
            pool = build_pool(fdset.file)
            infos = extract_method_infos(fdset.file, pool, b, compat_names=False)
            for mi in infos:
                tools[mi.tool_name()] = mi
            while len(self._fdps) <= b:
                self._fdps.append([])
                self.connections.append(self.connections[0])
                self.reflection_clients.append(None)
            self._fdps[b] = list(fdset.file)
        self.publish_tools(tools)

    # -- invocation: the hot string seam (discovery.go:346-369,
    #    reflection.go:333-391) — CPU reference path; the GPU engine replaces
    #    the two protojson transcodes with HIP kernels. ----------------------

    def _message_classes(self, mi: MethodInfo) -> Tuple[Any, Any]:
        key = id(mi)
        cached = self._msg_cls_cache.get(key)
        if cached is None:
            cached = (
                message_factory.GetMessageClass(mi.input_descriptor),
                message_factory.GetMessageClass(mi.output_descriptor),
            )
            self._msg_cls_cache[key] = cached
        return cached

    def encode_request(self, mi: MethodInfo, input_json: str) -> bytes:
        """JSON -> protobuf wire bytes (reference reflection.go:351-357)."""
        in_cls, _ = self._message_classes(mi)
        msg = json_format.Parse(input_json or "{}", in_cls())
        return msg.SerializeToString()

    def decode_response(self, mi: MethodInfo, wire: bytes) -> str:
        """protobuf wire bytes -> JSON (reference reflection.go:381)."""
        _, out_cls = self._message_classes(mi)
        msg = out_cls.FromString(wire)
        return json_format.MessageToJson(
            msg, indent=None, ensure_ascii=False, preserving_proto_field_name=False
        )

    def invoke_wire(
        self,
        mi: MethodInfo,
        request_wire: bytes,
        headers: Optional[Mapping[str, str]] = None,
        timeout_s: Optional[float] = None,
    ) -> bytes:
        """Raw wire-bytes unary invoke — the host-side I/O stage between the
        GPU encode and decode kernels (conn.Invoke, reflection.go:367-373)."""
        channel = self.connections[mi.backend_index].channel()
        callable_ = channel.unary_unary(
            mi.full_method_path,
            request_serializer=lambda b: b,
            response_deserializer=lambda b: b,
        )
        metadata = tuple((k.lower(), v) for k, v in (headers or {}).items())
        timeout = timeout_s if timeout_s is not None else self.config.grpc.request_timeout_s
        with self._count():
            return callable_(request_wire, metadata=metadata or None, timeout=timeout)

    def invoke_method_by_tool(
        self,
        tool_name: str,
        input_json: str,
        headers: Optional[Mapping[str, str]] = None,
        timeout_s: Optional[float] = None,
    ) -> str:
        """The full CPU hot path: JSON -> proto -> invoke -> proto -> JSON."""
        mi = self.get_method_by_tool(tool_name)
        if mi.is_streaming:
            raise ValueError(STREAMING_UNSUPPORTED_MSG)
        wire = self.encode_request(mi, input_json)
        out = self.invoke_wire(mi, wire, headers, timeout_s)
        return self.decode_response(mi, out)

    def invoke_streaming(
        self,
        tool_name: str,
        input_json: str,
        headers: Optional[Mapping[str, str]] = None,
        timeout_s: Optional[float] = None,
    ) -> Iterator[str]:
        """Server-streaming invocation (capability the reference rejects)."""
        mi = self.get_method_by_tool(tool_name)
        if mi.is_client_streaming:
            raise ValueError("client-streaming methods are not supported")
        if not mi.is_server_streaming:
            yield self.invoke_method_by_tool(tool_name, input_json, headers, timeout_s)
            return
        wire = self.encode_request(mi, input_json)
        channel = self.connections[mi.backend_index].channel()
        callable_ = channel.unary_stream(
            mi.full_method_path,
            request_serializer=lambda b: b,
            response_deserializer=lambda b: b,
        )
        metadata = tuple((k.lower(), v) for k, v in (headers or {}).items())
        timeout = timeout_s if timeout_s is not None else self.config.grpc.request_timeout_s
        with self._count():
            for chunk in callable_(wire, metadata=metadata or None, timeout=timeout):
                yield self.decode_response(mi, chunk)

    def invoke_streaming_wire(
        self,
        mi: MethodInfo,
        request_wire: bytes,
        headers: Optional[Mapping[str, str]] = None,
        timeout_s: Optional[float] = None,
    ) -> List[bytes]:
        """Server-streaming invoke returning raw wire chunks — the host I/O
        stage between the GPU encode kernel and a batched GPU decode of all
        stream messages (BASELINE config 4: per-stream GPU decode batches)."""
        channel = self.connections[mi.backend_index].channel()
        callable_ = channel.unary_stream(
            mi.full_method_path,
            request_serializer=lambda b: b,
            response_deserializer=lambda b: b,
        )
        metadata = tuple((k.lower(), v) for k, v in (headers or {}).items())
        timeout = timeout_s if timeout_s is not None else self.config.grpc.request_timeout_s
        with self._count():
            return list(callable_(request_wire, metadata=metadata or None, timeout=timeout))

    def _count(self):
        disc = self

        class _Ctr:
            def __enter__(self):
                return self

            def __exit__(self, et, ev, tb):
                with disc._lock:
                    disc._invocations += 1
                    if et is not None:
                        disc._errors += 1
                return False

        return _Ctr()

    # -- health / stats / reconnect (discovery.go:186-235, 278-333) -----------

    def health_check(self) -> bool:
        ok = all(c.health_check() for c in self.connections)
        if not ok:
            return False
        return all(rc is None or rc.health_check() for rc in self.reflection_clients)

    def reconnect_with_retry(self, attempts: int = 5, delay_s: float = 5.0) -> bool:
        """Reference discovery.go:186-235: bounded retry + full rediscovery."""
        for attempt in range(attempts):
            try:
                for i, conn in enumerate(self.connections):
                    channel = conn.reconnect()
                    self.reflection_clients[i] = ReflectionClient(channel)
                self.discover()
                return True
            except Exception as e:
                log.warning("reconnect attempt %d/%d failed: %s", attempt + 1, attempts, e)
                if attempt + 1 < attempts:
                    time.sleep(delay_s)
        return False

    def stats(self) -> Dict[str, Any]:
        tools = self._tools
        services = sorted({mi.service_name for mi in tools.values()})
        with self._lock:
            inv, errs = self._invocations, self._errors
        return {
            "serviceCount": len(services),
            "methodCount": len(tools),
            "isConnected": all(c.is_connected for c in self.connections),
            "services": services,
            "invocations": inv,
            "errors": errs,
            "toolsVersion": self._tools_version,
            "backends": [c.target for c in self.connections],
        }
