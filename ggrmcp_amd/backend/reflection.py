"""gRPC server-reflection client.

Re-design of the reference's ``pkg/grpc/reflection.go`` streaming reflection
client: ListServices (reflection.go:120-145), per-service
FileContainingSymbol with a FileDescriptorProto cache (reflection.go:196-254),
internal-service filtering (reflection.go:394-419), and MethodInfo
construction with input/output descriptor resolution via a temporary registry
with global fallback (reflection.go:296-330).

Speaks ``grpc.reflection.v1`` with automatic fallback to ``v1alpha`` (the
reference pins v1alpha; real servers commonly expose both).
"""

from __future__ import annotations

import logging
import threading
from typing import Dict, List, Optional, Sequence

import grpc
from google.protobuf import descriptor_pb2, descriptor_pool

from ..descriptors.loader import build_pool, extract_method_infos
from ..types import MethodInfo
from .reflection_proto import MESSAGES, V1, V1ALPHA

log = logging.getLogger("ggrmcp.reflection")

# reference reflection.go:394-419
INTERNAL_SERVICE_PREFIXES = (
    "grpc.reflection.",
    "grpc.health.",
    "grpc.channelz.",
    "grpc.testing.",
)


def filter_internal_services(services: Sequence[str]) -> List[str]:
    return [s for s in services if not s.startswith(INTERNAL_SERVICE_PREFIXES)]


class ReflectionError(RuntimeError):
    pass


class ReflectionClient:
    """Streaming reflection client (reference reflectionClient)."""

    def __init__(self, channel: grpc.Channel, timeout_s: float = 10.0) -> None:
        self._channel = channel
        self.timeout_s = timeout_s
        self._version: Optional[str] = None
        # FileDescriptorProto cache keyed by file name (reflection.go:32,196)
        self._fd_cache: Dict[str, descriptor_pb2.FileDescriptorProto] = {}
        self._lock = threading.Lock()

    # -- low-level stream round trip ----------------------------------------

    def _call(self, requests: list, version: str) -> list:
        msgs = MESSAGES[version]
        stub = self._channel.stream_stream(
            msgs.method_path,
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=msgs.ServerReflectionResponse.FromString,
        )
        responses = []
        stream = stub(iter(requests), timeout=self.timeout_s)
        for resp in stream:
            responses.append(resp)
            if len(responses) == len(requests):
                break
        if len(responses) != len(requests):
            raise ReflectionError(
                f"reflection stream returned {len(responses)}/{len(requests)} responses"
            )
        return responses

    def _roundtrip(self, requests: list) -> list:
        """Try v1 first, fall back to v1alpha, remember which worked."""
        versions = [self._version] if self._version else [V1, V1ALPHA]
        last_err: Optional[Exception] = None
        for version in versions:
            reqs = [self._convert(r, version) for r in requests]
            try:
                out = self._call(reqs, version)
                self._version = version
                return out
            except grpc.RpcError as e:
                code = e.code() if hasattr(e, "code") else None
                if code == grpc.StatusCode.UNIMPLEMENTED and self._version is None:
                    last_err = e
                    continue
                raise ReflectionError(f"reflection RPC failed: {e}") from e
        raise ReflectionError(f"no reflection service available: {last_err}")

    @staticmethod
    def _convert(request, version: str):
        msgs = MESSAGES[version]
        if request.DESCRIPTOR.full_name == f"{version}.ServerReflectionRequest":
            return request
        return msgs.ServerReflectionRequest.FromString(request.SerializeToString())

    @staticmethod
    def _check_error(resp) -> None:
        if resp.WhichOneof("message_response") == "error_response":
            err = resp.error_response
            raise ReflectionError(
                f"reflection error {err.error_code}: {err.error_message}"
            )

    # -- protocol operations (reflection.go:108-254) -------------------------

    def list_services(self) -> List[str]:
        req = MESSAGES[V1].ServerReflectionRequest(list_services="*")
        (resp,) = self._roundtrip([req])
        self._check_error(resp)
        return [s.name for s in resp.list_services_response.service]

    def files_containing_symbol(self, symbol: str) -> List[descriptor_pb2.FileDescriptorProto]:
        """Fetch (and cache) the files defining ``symbol`` + transitive deps
        the server chooses to send (reflection.go:196-254)."""
        req = MESSAGES[V1].ServerReflectionRequest(file_containing_symbol=symbol)
        (resp,) = self._roundtrip([req])
        self._check_error(resp)
        out = []
        with self._lock:
            for raw in resp.file_descriptor_response.file_descriptor_proto:
                fdp = descriptor_pb2.FileDescriptorProto.FromString(raw)
                if fdp.name not in self._fd_cache:
                    self._fd_cache[fdp.name] = fdp
                out.append(self._fd_cache[fdp.name])
        # resolve any dependencies the server did not include
        self._ensure_dependencies(out)
        return out

    def file_by_filename(self, name: str) -> descriptor_pb2.FileDescriptorProto:
        with self._lock:
            cached = self._fd_cache.get(name)
        if cached is not None:
            return cached
        req = MESSAGES[V1].ServerReflectionRequest(file_by_filename=name)
        (resp,) = self._roundtrip([req])
        self._check_error(resp)
        got = None
        with self._lock:
            for raw in resp.file_descriptor_response.file_descriptor_proto:
                fdp = descriptor_pb2.FileDescriptorProto.FromString(raw)
                self._fd_cache.setdefault(fdp.name, fdp)
                if fdp.name == name:
                    got = self._fd_cache[name]
        if got is None:
            raise ReflectionError(f"server did not return file {name}")
        return got

    def _ensure_dependencies(self, fdps: List[descriptor_pb2.FileDescriptorProto]) -> None:
        """Recursively fetch missing dependency files (skipping well-known
        types resolvable from the default pool)."""
        pending = list(fdps)
        seen = {f.name for f in fdps}
        while pending:
            fdp = pending.pop()
            for dep in fdp.dependency:
                if dep in seen:
                    continue
                seen.add(dep)
                with self._lock:
                    have = dep in self._fd_cache
                if have:
                    continue
                try:
                    descriptor_pool.Default().FindFileByName(dep)
                    continue  # well-known type; build_pool falls back to it
                except KeyError:
                    pass
                pending.append(self.file_by_filename(dep))

    # -- discovery (reflection.go:49-105, 257-330) ---------------------------

    def discover_methods(self, backend_index: int = 0) -> List[MethodInfo]:
        services = filter_internal_services(self.list_services())
        all_files: Dict[str, descriptor_pb2.FileDescriptorProto] = {}
        for svc in services:
            for fdp in self.files_containing_symbol(svc):
                all_files[fdp.name] = fdp
        with self._lock:
            # include cached dependency files fetched via _ensure_dependencies
            for name, fdp in self._fd_cache.items():
                all_files.setdefault(name, fdp)
        pool = build_pool(all_files.values())
        wanted = set(services)
        infos = [
            mi
            for mi in extract_method_infos(
                all_files.values(), pool, backend_index, compat_names=False
            )
            if mi.full_service_name in wanted or mi.service_name in wanted
        ]
        log.info(
            "reflection discovered %d methods across %d services (%s)",
            len(infos), len(services), self._version,
        )
        return infos

    def health_check(self) -> bool:
        """Reference reflection.go:439-451: ListServices with a timeout."""
        try:
            self.list_services()
            return True
        except Exception:
            return False
