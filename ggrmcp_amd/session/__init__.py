"""Session tracking (reference pkg/session)."""

from .manager import SessionContext, SessionManager  # noqa: F401
