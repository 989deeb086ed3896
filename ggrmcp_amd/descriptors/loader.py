"""FileDescriptorSet (.binpb) loading and descriptor-pool construction.

Re-design of the reference's ``pkg/descriptors/loader.go``:

* ``.binpb`` file -> ``FileDescriptorSet`` (loader.go:33-64);
* dependency-ordered pool construction with a global-registry fallback for
  well-known types (loader.go:67-134);
* walk services/methods -> MethodInfo including proto comment extraction from
  SourceCodeInfo (loader.go:137-216);
* the service-name compatibility shim truncating deep packages to their last
  segment, ``com.example.hello.HelloService`` -> ``hello.HelloService``
  (loader.go:219-235) — applied to tool naming only; the gRPC wire path keeps
  the full name (MethodInfo.full_service_name).

``build_pool``/``extract_comments`` are shared with the reflection client
(ggrmcp_amd/backend/reflection.py), which receives the same
FileDescriptorProtos over the wire instead of from a file.
"""

from __future__ import annotations

import logging
from typing import Dict, Iterable, List, Optional, Tuple

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

# Importing the well-known-type modules registers their files in the default
# descriptor pool, which is the fallback registry for dependency resolution
# (the reference's protoregistry.GlobalFiles fallback, loader.go:96-110).
from google.protobuf import (  # noqa: F401
    any_pb2,
    api_pb2,
    duration_pb2,
    empty_pb2,
    field_mask_pb2,
    source_context_pb2,
    struct_pb2,
    timestamp_pb2,
    type_pb2,
    wrappers_pb2,
)

from ..types import MethodInfo, SourceLocation

log = logging.getLogger("ggrmcp.descriptors")


def build_pool(
    fdps: Iterable[descriptor_pb2.FileDescriptorProto],
    pool: Optional[descriptor_pool.DescriptorPool] = None,
) -> descriptor_pool.DescriptorPool:
    """Add files to a private pool in dependency order (loader.go:67-134).

    Missing dependencies (well-known types like google/protobuf/*.proto) are
    pulled from the process-global default pool — the reference's
    global-registry fallback.
    """
    pool = pool or descriptor_pool.DescriptorPool()
    by_name: Dict[str, descriptor_pb2.FileDescriptorProto] = {f.name: f for f in fdps}
    done: Dict[str, bool] = {}

    def ensure(name: str, stack: Tuple[str, ...] = ()) -> None:
        if done.get(name):
            return
        if name in stack:
            raise ValueError(f"circular proto dependency involving {name}")
        try:
            pool.FindFileByName(name)
            done[name] = True
            return
        except KeyError:
            pass
        fdp = by_name.get(name)
        if fdp is None:
            # global-registry fallback (well-known types)
            default = descriptor_pool.Default().FindFileByName(name)
            copy = descriptor_pb2.FileDescriptorProto()
            default.CopyToProto(copy)
            for dep in copy.dependency:
                ensure(dep, stack + (name,))
            pool.Add(copy)
            done[name] = True
            return
        for dep in fdp.dependency:
            ensure(dep, stack + (name,))
        pool.Add(fdp)
        done[name] = True

    for name in by_name:
        ensure(name)
    return pool


def extract_comments(
    fdp: descriptor_pb2.FileDescriptorProto,
) -> Dict[Tuple[int, ...], Tuple[str, str]]:
    """SourceCodeInfo path -> (leading, trailing) comments (loader.go:195-216).

    Proto path convention: service i = (6, i); method j of service i =
    (6, i, 2, j); message i = (4, i).
    """
    out: Dict[Tuple[int, ...], Tuple[str, str]] = {}
    for loc in fdp.source_code_info.location:
        if loc.leading_comments or loc.trailing_comments:
            out[tuple(loc.path)] = (
                loc.leading_comments.strip(),
                loc.trailing_comments.strip(),
            )
    return out


def compat_service_name(full_name: str) -> str:
    """Reference loader.go:219-235: keep at most the last package segment +
    service name (``a.b.c.Svc`` -> ``c.Svc``)."""
    parts = full_name.split(".")
    if len(parts) <= 2:
        return full_name
    return ".".join(parts[-2:])


def extract_method_infos(
    fdps: Iterable[descriptor_pb2.FileDescriptorProto],
    pool: descriptor_pool.DescriptorPool,
    backend_index: int = 0,
    compat_names: bool = True,
) -> List[MethodInfo]:
    """Walk every service of every file -> MethodInfo (loader.go:137-192)."""
    infos: List[MethodInfo] = []
    for fdp in fdps:
        comments = extract_comments(fdp)
        pkg = fdp.package
        for si, svc in enumerate(fdp.service):
            full_service = f"{pkg}.{svc.name}" if pkg else svc.name
            display = compat_service_name(full_service) if compat_names else full_service
            for mi, method in enumerate(svc.method):
                in_desc = pool.FindMessageTypeByName(method.input_type.lstrip("."))
                out_desc = pool.FindMessageTypeByName(method.output_type.lstrip("."))
                lead, trail = comments.get((6, si, 2, mi), ("", ""))
                desc = lead or trail
                infos.append(
                    MethodInfo(
                        service_name=display,
                        full_service_name=full_service,
                        method_name=method.name,
                        input_descriptor=in_desc,
                        output_descriptor=out_desc,
                        is_client_streaming=method.client_streaming,
                        is_server_streaming=method.server_streaming,
                        description=desc,
                        source=SourceLocation(file=fdp.name),
                        backend_index=backend_index,
                    )
                )
    return infos


class DescriptorLoader:
    """Loads a compiled FileDescriptorSet (reference descriptors.Loader)."""

    def __init__(self, path: str) -> None:
        self.path = path
        self.fdset = descriptor_pb2.FileDescriptorSet()
        self.pool: Optional[descriptor_pool.DescriptorPool] = None

    def load(self) -> "DescriptorLoader":
        """Reference LoadFromFile (loader.go:33-64)."""
        with open(self.path, "rb") as fh:
            data = fh.read()
        if not data:
            raise ValueError(f"descriptor set {self.path} is empty")
        self.fdset.ParseFromString(data)
        if not self.fdset.file:
            raise ValueError(f"descriptor set {self.path} contains no files")
        return self

    def build_registry(self) -> descriptor_pool.DescriptorPool:
        """Reference BuildRegistry (loader.go:67-134)."""
        self.pool = build_pool(self.fdset.file)
        return self.pool

    def extract_method_info(self, backend_index: int = 0) -> List[MethodInfo]:
        """Reference ExtractMethodInfo (loader.go:137-192)."""
        if self.pool is None:
            self.build_registry()
        return extract_method_infos(self.fdset.file, self.pool, backend_index)

    def message_class(self, full_name: str):
        if self.pool is None:
            self.build_registry()
        return message_factory.GetMessageClass(self.pool.FindMessageTypeByName(full_name))
