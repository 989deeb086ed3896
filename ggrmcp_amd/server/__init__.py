"""HTTP/JSON-RPC surface (reference pkg/server)."""

from .handler import MCPHandler  # noqa: F401
from .middleware import (  # noqa: F401
    Request,
    Response,
    chain_middleware,
    default_middleware,
)
