"""gRPC backend layer: connection, reflection discovery, invocation
(reference pkg/grpc)."""

from .connection import ConnectionManager  # noqa: F401
from .discovery import ServiceDiscoverer  # noqa: F401
from .reflection import ReflectionClient  # noqa: F401
