"""Shared record types flowing through every layer.

MI355X-native re-design of the reference's ``pkg/types/service.go``
(MethodInfo: service.go:15-43, GenerateToolName: service.go:53-61,
SourceLocation: service.go:64-67).  Unlike the Go reference, MethodInfo here
also carries the index of the backend it was discovered on (multi-backend /
centralized-gateway mode) and a stable shard key used by the DP session
sharder.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

from google.protobuf.descriptor import Descriptor


@dataclass
class SourceLocation:
    """Proto source location of a method/message (service.go:64-67)."""

    file: str = ""
    line: int = 0


@dataclass
class MethodInfo:
    """The single record describing one callable gRPC method.

    Mirrors reference pkg/types/service.go:15-43: full service name
    (``package.Service``), method name, resolved input/output message
    descriptors, streaming flags and extracted proto comments.
    """

    service_name: str
    method_name: str
    input_descriptor: Optional[Descriptor] = None
    output_descriptor: Optional[Descriptor] = None
    is_client_streaming: bool = False
    is_server_streaming: bool = False
    description: str = ""
    input_comment: str = ""
    output_comment: str = ""
    source: SourceLocation = field(default_factory=SourceLocation)
    # MI355X additions: which backend (centralized-gateway mode) owns the
    # method, used by the invoker; reference has exactly one backend.
    backend_index: int = 0
    # When the compat shim truncates service_name for tool naming (reference
    # loader.go:219-235), the full name still needed for the gRPC wire path.
    full_service_name: str = ""

    @property
    def full_method_path(self) -> str:
        """gRPC wire path ``/package.Service/Method``."""
        return f"/{self.full_service_name or self.service_name}/{self.method_name}"

    @property
    def is_streaming(self) -> bool:
        return self.is_client_streaming or self.is_server_streaming

    def tool_name(self) -> str:
        """MCP tool name (reference service.go:53-61).

        lowercase(service) with dots replaced by underscores, ``_``,
        lowercase(method): ``hello.HelloService/SayHello`` ->
        ``hello_helloservice_sayhello``.
        """
        return generate_tool_name(self.service_name, self.method_name)


def generate_tool_name(service_name: str, method_name: str) -> str:
    """Reference pkg/types/service.go:53-61."""
    svc = service_name.lower().replace(".", "_")
    return f"{svc}_{method_name.lower()}"
