"""gRPC server-reflection protocol messages, built programmatically.

The environment has no ``grpc_reflection`` package and no ``protoc``, so the
reflection protocol's messages (``grpc.reflection.v1alpha`` and the identical
``grpc.reflection.v1``) are constructed here as ``FileDescriptorProto``s and
compiled into a private descriptor pool.  The wire format is identical to the
canonical ``reflection.proto`` used by the reference's Go client
(/root/reference/pkg/grpc/reflection.go:120-145 drives the same RPCs).

Both the client (ggrmcp_amd/backend/reflection.py) and the in-process test /
example servers (ggrmcp_amd/backend/reflection_server.py) use these classes.
"""

from __future__ import annotations

from typing import Dict

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_F = descriptor_pb2.FieldDescriptorProto

V1ALPHA = "grpc.reflection.v1alpha"
V1 = "grpc.reflection.v1"


def _add_field(msg, name, number, ftype, label=_F.LABEL_OPTIONAL, type_name=None, oneof=None):
    f = msg.field.add()
    f.name = name
    f.number = number
    f.type = ftype
    f.label = label
    if type_name is not None:
        f.type_name = type_name
    if oneof is not None:
        f.oneof_index = oneof


def _build_file(package: str) -> descriptor_pb2.FileDescriptorProto:
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = package.replace(".", "/") + "/reflection.proto"
    fdp.package = package
    fdp.syntax = "proto3"
    p = "." + package

    req = fdp.message_type.add()
    req.name = "ServerReflectionRequest"
    req.oneof_decl.add().name = "message_request"
    _add_field(req, "host", 1, _F.TYPE_STRING)
    _add_field(req, "file_by_filename", 3, _F.TYPE_STRING, oneof=0)
    _add_field(req, "file_containing_symbol", 4, _F.TYPE_STRING, oneof=0)
    _add_field(
        req, "file_containing_extension", 5, _F.TYPE_MESSAGE,
        type_name=f"{p}.ExtensionRequest", oneof=0,
    )
    _add_field(req, "all_extension_numbers_of_type", 6, _F.TYPE_STRING, oneof=0)
    _add_field(req, "list_services", 7, _F.TYPE_STRING, oneof=0)

    ext = fdp.message_type.add()
    ext.name = "ExtensionRequest"
    _add_field(ext, "containing_type", 1, _F.TYPE_STRING)
    _add_field(ext, "extension_number", 2, _F.TYPE_INT32)

    resp = fdp.message_type.add()
    resp.name = "ServerReflectionResponse"
    resp.oneof_decl.add().name = "message_response"
    _add_field(resp, "valid_host", 1, _F.TYPE_STRING)
    _add_field(
        resp, "original_request", 2, _F.TYPE_MESSAGE,
        type_name=f"{p}.ServerReflectionRequest",
    )
    _add_field(
        resp, "file_descriptor_response", 4, _F.TYPE_MESSAGE,
        type_name=f"{p}.FileDescriptorResponse", oneof=0,
    )
    _add_field(
        resp, "all_extension_numbers_response", 5, _F.TYPE_MESSAGE,
        type_name=f"{p}.ExtensionNumberResponse", oneof=0,
    )
    _add_field(
        resp, "list_services_response", 6, _F.TYPE_MESSAGE,
        type_name=f"{p}.ListServiceResponse", oneof=0,
    )
    _add_field(
        resp, "error_response", 7, _F.TYPE_MESSAGE,
        type_name=f"{p}.ErrorResponse", oneof=0,
    )

    fdr = fdp.message_type.add()
    fdr.name = "FileDescriptorResponse"
    _add_field(fdr, "file_descriptor_proto", 1, _F.TYPE_BYTES, label=_F.LABEL_REPEATED)

    enr = fdp.message_type.add()
    enr.name = "ExtensionNumberResponse"
    _add_field(enr, "base_type_name", 1, _F.TYPE_STRING)
    _add_field(enr, "extension_number", 2, _F.TYPE_INT32, label=_F.LABEL_REPEATED)

    lsr = fdp.message_type.add()
    lsr.name = "ListServiceResponse"
    _add_field(lsr, "service", 1, _F.TYPE_MESSAGE, label=_F.LABEL_REPEATED,
               type_name=f"{p}.ServiceResponse")

    svc_resp = fdp.message_type.add()
    svc_resp.name = "ServiceResponse"
    _add_field(svc_resp, "name", 1, _F.TYPE_STRING)

    err = fdp.message_type.add()
    err.name = "ErrorResponse"
    _add_field(err, "error_code", 1, _F.TYPE_INT32)
    _add_field(err, "error_message", 2, _F.TYPE_STRING)

    svc = fdp.service.add()
    svc.name = "ServerReflection"
    m = svc.method.add()
    m.name = "ServerReflectionInfo"
    m.input_type = f"{p}.ServerReflectionRequest"
    m.output_type = f"{p}.ServerReflectionResponse"
    m.client_streaming = True
    m.server_streaming = True
    return fdp


_pool = descriptor_pool.DescriptorPool()
_pool.Add(_build_file(V1ALPHA))
_pool.Add(_build_file(V1))


class _Messages:
    """Message classes for one reflection package version."""

    def __init__(self, package: str) -> None:
        self.package = package
        get = lambda n: message_factory.GetMessageClass(  # noqa: E731
            _pool.FindMessageTypeByName(f"{package}.{n}")
        )
        self.ServerReflectionRequest = get("ServerReflectionRequest")
        self.ServerReflectionResponse = get("ServerReflectionResponse")
        self.ExtensionRequest = get("ExtensionRequest")
        self.FileDescriptorResponse = get("FileDescriptorResponse")
        self.ListServiceResponse = get("ListServiceResponse")
        self.ServiceResponse = get("ServiceResponse")
        self.ErrorResponse = get("ErrorResponse")

    @property
    def method_path(self) -> str:
        return f"/{self.package}.ServerReflection/ServerReflectionInfo"


MESSAGES: Dict[str, _Messages] = {V1ALPHA: _Messages(V1ALPHA), V1: _Messages(V1)}
