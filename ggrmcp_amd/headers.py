"""Security-filtered header forwarding.

Re-design of the reference's ``pkg/headers/filter.go`` (Filter: filter.go:10-18,
ShouldForward/FilterHeaders: filter.go:28-68) with the same semantics:

* forwarding disabled  -> empty map (filter.go:66-68)
* blocked list always wins over the allow list (filter.go:33-42)
* ``forward_all`` escape hatch, still subject to the blocked list
  (filter.go:44-47)
* otherwise allow-list only (filter.go:49-58)
* case-insensitive by default (filter.go:28-31)

Defaults match reference pkg/config/config.go:246-269.
"""

from __future__ import annotations

from typing import Dict, Iterable, Mapping

DEFAULT_ALLOWED = (
    "authorization",
    "x-trace-id",
    "x-user-id",
    "x-request-id",
    "user-agent",
    "x-forwarded-for",
    "x-real-ip",
)

DEFAULT_BLOCKED = (
    "cookie",
    "set-cookie",
    "host",
    "content-length",
    "content-type",
    "connection",
    "upgrade",
    "mcp-session-id",
)


class HeaderFilter:
    """Pure-function header filter (reference pkg/headers/filter.go)."""

    def __init__(
        self,
        enabled: bool = True,
        allowed: Iterable[str] = DEFAULT_ALLOWED,
        blocked: Iterable[str] = DEFAULT_BLOCKED,
        forward_all: bool = False,
        case_insensitive: bool = True,
    ) -> None:
        self.enabled = enabled
        self.forward_all = forward_all
        self.case_insensitive = case_insensitive
        norm = (lambda s: s.lower()) if case_insensitive else (lambda s: s)
        self._norm = norm
        self.allowed = {norm(h) for h in allowed}
        self.blocked = {norm(h) for h in blocked}

    @classmethod
    def from_config(cls, cfg) -> "HeaderFilter":
        """Build from a config.HeaderForwardingConfig."""
        return cls(
            enabled=cfg.enabled,
            allowed=cfg.allowed_headers,
            blocked=cfg.blocked_headers,
            forward_all=cfg.forward_all,
            case_insensitive=cfg.case_insensitive,
        )

    def should_forward(self, name: str) -> bool:
        """Reference filter.go:28-58."""
        if not self.enabled:
            return False
        key = self._norm(name)
        if key in self.blocked:
            return False
        if self.forward_all:
            return True
        return key in self.allowed

    def filter_headers(self, headers: Mapping[str, str]) -> Dict[str, str]:
        """Reference filter.go:61-93: returns only forwardable headers."""
        if not self.enabled:
            return {}
        return {k: v for k, v in headers.items() if self.should_forward(k)}
