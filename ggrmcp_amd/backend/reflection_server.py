"""In-process gRPC reflection *server* support.

The reference consumes reflection from Go backends that get it for free from
``grpc/reflection``; this environment has no ``grpc_reflection`` package, so
example backends and tests register this generic handler instead.  Serves
both ``grpc.reflection.v1`` and ``v1alpha`` (the reference client pins
v1alpha, reflection.go:120).
"""

from __future__ import annotations

from typing import Dict, Iterable, List

import grpc
from google.protobuf import descriptor_pb2

from .reflection_proto import MESSAGES, V1, V1ALPHA


class ReflectionServicer:
    """Answers ServerReflectionInfo streams from a set of files + services."""

    def __init__(
        self,
        service_names: Iterable[str],
        fdps: Iterable[descriptor_pb2.FileDescriptorProto],
    ) -> None:
        self.service_names = list(service_names)
        self.files: Dict[str, descriptor_pb2.FileDescriptorProto] = {
            f.name: f for f in fdps
        }
        self._symbol_to_file: Dict[str, str] = {}
        for f in self.files.values():
            pkg = f.package
            prefix = pkg + "." if pkg else ""
            for svc in f.service:
                self._symbol_to_file[prefix + svc.name] = f.name
                for m in svc.method:
                    self._symbol_to_file[f"{prefix}{svc.name}.{m.name}"] = f.name
            for msg in f.message_type:
                self._symbol_to_file[prefix + msg.name] = f.name
            for en in f.enum_type:
                self._symbol_to_file[prefix + en.name] = f.name

    def _transitive(self, name: str) -> List[bytes]:
        """File + transitive deps we own, serialized (dep-first order)."""
        out: List[bytes] = []
        seen = set()

        def walk(n: str) -> None:
            if n in seen or n not in self.files:
                return
            seen.add(n)
            fdp = self.files[n]
            for dep in fdp.dependency:
                walk(dep)
            out.append(fdp.SerializeToString())

        walk(name)
        return out

    def _handle(self, request, msgs):
        resp = msgs.ServerReflectionResponse()
        resp.original_request.CopyFrom(
            msgs.ServerReflectionRequest.FromString(request.SerializeToString())
        )
        which = request.WhichOneof("message_request")
        if which == "list_services":
            for name in self.service_names:
                resp.list_services_response.service.add().name = name
        elif which == "file_containing_symbol":
            fname = self._symbol_to_file.get(request.file_containing_symbol)
            if fname is None:
                resp.error_response.error_code = grpc.StatusCode.NOT_FOUND.value[0]
                resp.error_response.error_message = (
                    f"symbol not found: {request.file_containing_symbol}"
                )
            else:
                resp.file_descriptor_response.file_descriptor_proto.extend(
                    self._transitive(fname)
                )
        elif which == "file_by_filename":
            if request.file_by_filename in self.files:
                resp.file_descriptor_response.file_descriptor_proto.extend(
                    self._transitive(request.file_by_filename)
                )
            else:
                resp.error_response.error_code = grpc.StatusCode.NOT_FOUND.value[0]
                resp.error_response.error_message = (
                    f"file not found: {request.file_by_filename}"
                )
        else:
            resp.error_response.error_code = grpc.StatusCode.UNIMPLEMENTED.value[0]
            resp.error_response.error_message = f"unsupported request: {which}"
        return resp

    def _stream(self, msgs):
        def handler(request_iterator, context):
            for request in request_iterator:
                yield self._handle(request, msgs)

        return handler

    def generic_handlers(self) -> List[grpc.GenericRpcHandler]:
        handlers = []
        for version in (V1, V1ALPHA):
            msgs = MESSAGES[version]
            method = grpc.stream_stream_rpc_method_handler(
                self._stream(msgs),
                request_deserializer=msgs.ServerReflectionRequest.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            )
            handlers.append(
                grpc.method_handlers_generic_handler(
                    f"{version}.ServerReflection",
                    {"ServerReflectionInfo": method},
                )
            )
        return handlers


def enable_reflection(server: grpc.Server, service_names, fdps) -> None:
    servicer = ReflectionServicer(service_names, fdps)
    server.add_generic_rpc_handlers(tuple(servicer.generic_handlers()))
