"""gRPC channel management.

Re-design of the reference's ``pkg/grpc/connection.go``: one channel per
backend with keepalive (connection.go:47-54), 4 MB message caps
(connection.go:55-57), bounded connect (connection.go:60-67), a connectivity
health check (connection.go:116-142) and reconnect (connection.go:103-106).

Unlike the reference (exactly one backend), a gateway process may hold
several ConnectionManagers — centralized-gateway mode (reference
README.md:129-141 documents the pattern without wiring it).
"""

from __future__ import annotations

import logging
import threading
from typing import Optional

import grpc

from ..config import ConnectionConfig

log = logging.getLogger("ggrmcp.connection")


class ConnectionError_(RuntimeError):
    pass


class ConnectionManager:
    """Owns one grpc.Channel to one backend (reference connectionManager)."""

    def __init__(self, cfg: Optional[ConnectionConfig] = None, target: Optional[str] = None):
        self.cfg = cfg or ConnectionConfig()
        self._target = target or self.cfg.target
        self._channel: Optional[grpc.Channel] = None
        self._lock = threading.RLock()
        self._connected = False

    @property
    def target(self) -> str:
        return self._target

    def connect(self, timeout_s: Optional[float] = None) -> grpc.Channel:
        """Dial + block until READY (reference connection.go:34-67)."""
        timeout = timeout_s if timeout_s is not None else self.cfg.connect_timeout_s
        with self._lock:
            if self._channel is not None and self._connected:
                return self._channel
            options = [
                ("grpc.keepalive_time_ms", int(self.cfg.keepalive_time_s * 1000)),
                ("grpc.keepalive_timeout_ms", int(self.cfg.keepalive_timeout_s * 1000)),
                ("grpc.keepalive_permit_without_calls", int(self.cfg.permit_without_stream)),
                ("grpc.max_send_message_length", self.cfg.max_send_msg_bytes),
                ("grpc.max_receive_message_length", self.cfg.max_recv_msg_bytes),
            ]
            channel = grpc.insecure_channel(self._target, options=options)
            fut = grpc.channel_ready_future(channel)
            try:
                fut.result(timeout=timeout)
            except grpc.FutureTimeoutError:
                fut.cancel()
                channel.close()
                raise ConnectionError_(
                    f"failed to connect to gRPC backend {self._target} within {timeout}s"
                )
            finally:
                fut.cancel()  # stop the connectivity poll thread
            self._channel = channel
            self._connected = True
            log.info("connected to gRPC backend %s", self._target)
            return channel

    def channel(self) -> grpc.Channel:
        with self._lock:
            if self._channel is None:
                raise ConnectionError_("not connected")
            return self._channel

    @property
    def is_connected(self) -> bool:
        with self._lock:
            return self._connected

    def health_check(self, timeout_s: float = 5.0) -> bool:
        """Connectivity-state health check (reference connection.go:116-142)."""
        with self._lock:
            channel = self._channel
        if channel is None:
            return False
        try:
            # connectivity-state probe without spawning a poll thread
            # (grpc.channel_ready_future leaves a poller racing close())
            state = channel._channel.check_connectivity_state(True)
            ready = grpc.ChannelConnectivity.READY.value[0]
            idle = grpc.ChannelConnectivity.IDLE.value[0]
            if state == ready:
                return True
            if state == idle:
                # IDLE channels flip to READY on first use; treat as healthy
                # (the reference's WaitForStateChange loop does the same walk)
                return True
            deadline = timeout_s
            import time as _time

            step = 0.05
            while deadline > 0:
                _time.sleep(step)
                deadline -= step
                if channel._channel.check_connectivity_state(True) == ready:
                    return True
            return False
        except Exception:
            fut = grpc.channel_ready_future(channel)
            try:
                fut.result(timeout=timeout_s)
                return True
            except Exception:
                return False
            finally:
                fut.cancel()

    def reconnect(self, timeout_s: Optional[float] = None) -> grpc.Channel:
        """Reference connection.go:103-106 (Reconnect = close + Connect)."""
        self.close()
        return self.connect(timeout_s)

    def close(self) -> None:
        with self._lock:
            if self._channel is not None:
                try:
                    self._channel.close()
                finally:
                    self._channel = None
                    self._connected = False
