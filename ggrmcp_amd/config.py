"""Configuration tree.

Re-design of the reference's ``pkg/config/config.go`` (Config struct:
config.go:9-208, Default(): 211-312, Development(): 315-325, Validate():
328-357).  Two deliberate improvements over the reference:

* the tree is actually loadable from YAML/JSON files (the reference declares
  yaml/json tags but never loads a file — SURVEY.md §2 component 2), and
* a ``gpu`` section describes the MI355X batch-engine topology (device count,
  session shard policy, HIP stream depth, batch window) which has no
  reference equivalent (the reference is CPU-only Go).

Defaults match the reference's ``Default()`` values where they exist
(config.go:211-312).
"""

from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


@dataclass
class ConnectionConfig:
    """Reference config.go GRPC connection section + connection.go:47-58."""

    host: str = "localhost"
    port: int = 50051
    # keepalive (reference connection.go:47-54)
    keepalive_time_s: float = 10.0
    keepalive_timeout_s: float = 5.0
    permit_without_stream: bool = True
    connect_timeout_s: float = 5.0
    max_send_msg_bytes: int = 4 * 1024 * 1024  # connection.go:55-57
    max_recv_msg_bytes: int = 4 * 1024 * 1024
    request_timeout_s: float = 30.0  # config.go:235
    use_tls: bool = False
    uds: str = ""  # unix-domain socket path (overrides host:port)
    # h2 connections per backend for the native batch transport.  8 suits
    # ~1 KB payloads; raise to 16 for multi-MB/step workloads, which are
    # UDS-syscall bound (profiles/wide64_conns.log: +27% at 64 KB payloads)
    native_connections: int = 8

    @property
    def target(self) -> str:
        if self.uds:
            return f"unix:{self.uds}"
        return f"{self.host}:{self.port}"


@dataclass
class DescriptorSetConfig:
    """Reference config.go DescriptorSetConfig."""

    enabled: bool = False
    path: str = ""
    prefer_over_reflection: bool = False
    include_source_info: bool = True


@dataclass
class HeaderForwardingConfig:
    """Reference config.go:246-269."""

    enabled: bool = True
    allowed_headers: List[str] = field(
        default_factory=lambda: [
            "authorization",
            "x-trace-id",
            "x-user-id",
            "x-request-id",
            "user-agent",
            "x-forwarded-for",
            "x-real-ip",
        ]
    )
    blocked_headers: List[str] = field(
        default_factory=lambda: [
            "cookie",
            "set-cookie",
            "host",
            "content-length",
            "content-type",
            "connection",
            "upgrade",
            "mcp-session-id",
        ]
    )
    forward_all: bool = False
    case_insensitive: bool = True


@dataclass
class SessionConfig:
    """Reference session/manager.go:53-66 + config.go:289-294."""

    ttl_s: float = 30 * 60.0
    cleanup_interval_s: float = 5 * 60.0
    max_sessions: int = 10_000
    rate_limit_per_min: int = 100
    rate_limit_burst: int = 20
    rate_limit_enabled: bool = False  # ref default stack never calls it
    # /dev/shm path for the C++ shared-memory session table; when set, every
    # serve_dp rank on the port maps the SAME table, so a session keeps its
    # rate-limit/block/call-count state no matter which rank the kernel's
    # SO_REUSEPORT balancing lands its reconnect on.  Empty = per-process.
    shared_table_path: str = ""


@dataclass
class ServerConfig:
    """Reference cmd/grmcp/main.go:202-208 + middleware defaults."""

    http_port: int = 50053  # config.Default() value (config.go:214)
    read_timeout_s: float = 15.0
    write_timeout_s: float = 15.0
    idle_timeout_s: float = 60.0
    handler_timeout_s: float = 30.0  # handler.go:239
    shutdown_drain_s: float = 30.0  # main.go:94-112
    max_body_bytes: int = 1024 * 1024  # middleware.go:288
    max_response_bytes: int = 16 * 1024 * 1024  # config.go:283
    rate_limit_rps: float = 100.0  # middleware.go:286
    rate_limit_burst: int = 200
    rate_limit_enabled: bool = True
    cors_enabled: bool = True
    # SO_REUSEPORT on the native frontend: run one gateway process per GPU
    # rank all bound to the SAME port; the kernel balances connections
    # (tools/serve_dp.py).  No reference equivalent (single process).
    reuse_port: bool = False
    # native frontend epoll reactor shards (HTTP parse + socket I/O);
    # a single reactor measured ~25 us/request of parse+epoll+write
    reactors: int = 4
    security_headers_enabled: bool = True


@dataclass
class GPUConfig:
    """MI355X batch-engine topology — no reference equivalent."""

    enabled: bool = True
    devices: int = 1  # engines (== visible GPUs) for DP session sharding
    batch_window_us: int = 200  # adaptive batch collection window
    max_batch: int = 4096  # requests per GPU batch
    # Engine instances (each with its own HIP stream + arenas) per device.
    # 2 measured best with the native span executor: overlap pays, but HIP
    # runtime contention grows superlinearly with instance count
    # (profiles/streams_sweep: 1->207k, 2->245k, 4->233k req/s same box)
    streams: int = 2
    max_request_bytes: int = 1024 * 1024  # per-request staging cap
    pinned_pool_bytes: int = 256 * 1024 * 1024  # pinned host staging pool
    device_pool_bytes: int = 1024 * 1024 * 1024  # HBM arena per engine
    require_gpu: bool = False  # fail loudly instead of CPU fallback


@dataclass
class LoggingConfig:
    level: str = "info"
    development: bool = False


@dataclass
class Config:
    """Root config (reference config.go:9-27)."""

    grpc: ConnectionConfig = field(default_factory=ConnectionConfig)
    # centralized-gateway mode: additional backends beyond `grpc`
    # (reference README.md:129-141 documents the pattern; code wires one).
    extra_backends: List[ConnectionConfig] = field(default_factory=list)
    descriptor_set: DescriptorSetConfig = field(default_factory=DescriptorSetConfig)
    header_forwarding: HeaderForwardingConfig = field(default_factory=HeaderForwardingConfig)
    session: SessionConfig = field(default_factory=SessionConfig)
    server: ServerConfig = field(default_factory=ServerConfig)
    gpu: GPUConfig = field(default_factory=GPUConfig)
    logging: LoggingConfig = field(default_factory=LoggingConfig)

    # ---- profiles ---------------------------------------------------------

    @classmethod
    def default(cls) -> "Config":
        """Reference Default() (config.go:211-312)."""
        return cls()

    @classmethod
    def development(cls) -> "Config":
        """Reference Development() (config.go:315-325)."""
        cfg = cls()
        cfg.logging.level = "debug"
        cfg.logging.development = True
        cfg.server.rate_limit_enabled = False
        return cfg

    # ---- file loading (improvement over reference) ------------------------

    @classmethod
    def from_dict(cls, data: Dict[str, Any]) -> "Config":
        cfg = cls()
        _merge_dataclass(cfg, data)
        return cfg

    @classmethod
    def from_file(cls, path: str) -> "Config":
        with open(path, "r", encoding="utf-8") as fh:
            text = fh.read()
        if path.endswith((".yaml", ".yml")):
            import yaml

            data = yaml.safe_load(text) or {}
        else:
            data = json.loads(text)
        if not isinstance(data, dict):
            raise ValueError(f"config file {path}: top level must be a mapping")
        return cls.from_dict(data)

    def to_dict(self) -> Dict[str, Any]:
        return dataclasses.asdict(self)

    # ---- validation (reference Validate(), config.go:328-357) -------------

    def validate(self) -> None:
        errs: List[str] = []
        if not (0 < self.grpc.port < 65536):
            errs.append(f"grpc.port out of range: {self.grpc.port}")
        if not (0 <= self.server.http_port < 65536):  # 0 = auto-assign
            errs.append(f"server.http_port out of range: {self.server.http_port}")
        if not self.grpc.host:
            errs.append("grpc.host empty")
        if self.grpc.request_timeout_s <= 0:
            errs.append("grpc.request_timeout_s must be > 0")
        if self.session.max_sessions <= 0:
            errs.append("session.max_sessions must be > 0")
        if self.server.max_body_bytes <= 0:
            errs.append("server.max_body_bytes must be > 0")
        if self.descriptor_set.enabled and not self.descriptor_set.path:
            errs.append("descriptor_set.enabled but descriptor_set.path empty")
        if self.gpu.enabled:
            if self.gpu.devices < 1 or self.gpu.devices > 8:
                errs.append(f"gpu.devices out of range [1,8]: {self.gpu.devices}")
            if self.gpu.max_batch < 1:
                errs.append("gpu.max_batch must be >= 1")
            if self.gpu.streams < 1:
                errs.append("gpu.streams must be >= 1")
        if self.grpc.native_connections < 1:
            errs.append("grpc.native_connections must be >= 1")
        if self.logging.level not in ("debug", "info", "warn", "warning", "error"):
            errs.append(f"unknown logging.level: {self.logging.level}")
        if errs:
            raise ValueError("invalid config: " + "; ".join(errs))

    def all_backends(self) -> List[ConnectionConfig]:
        return [self.grpc] + list(self.extra_backends)


def _merge_dataclass(obj: Any, data: Dict[str, Any]) -> None:
    """Recursively merge a plain dict onto a dataclass instance."""
    names = {f.name: f for f in dataclasses.fields(obj)}
    for key, value in data.items():
        if key not in names:
            raise ValueError(f"unknown config key: {key!r}")
        current = getattr(obj, key)
        if dataclasses.is_dataclass(current) and isinstance(value, dict):
            _merge_dataclass(current, value)
        elif key == "extra_backends" and isinstance(value, list):
            backends = []
            for item in value:
                bc = ConnectionConfig()
                _merge_dataclass(bc, item)
                backends.append(bc)
            setattr(obj, key, backends)
        else:
            setattr(obj, key, value)
