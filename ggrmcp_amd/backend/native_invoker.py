"""Native gRPC wire client (the host I/O stage of the GPU pipeline).

Wraps the C++ _h2grpc batch client (ops/csrc/h2grpc.cpp: nghttp2 h2c,
multiplexed streams, GIL-released batch submission) behind a small Python
API and a grpc.RpcError-compatible error type.  This replaces the grpcio
per-call stub for the batched hot path — the same role conn.Invoke plays in
the reference (reflection.go:367-373), at ~50x the call rate.
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional, Sequence, Tuple

log = logging.getLogger("ggrmcp.native_invoker")

_CODE_NAMES = [
    "OK", "CANCELLED", "UNKNOWN", "INVALID_ARGUMENT", "DEADLINE_EXCEEDED",
    "NOT_FOUND", "ALREADY_EXISTS", "PERMISSION_DENIED", "RESOURCE_EXHAUSTED",
    "FAILED_PRECONDITION", "ABORTED", "OUT_OF_RANGE", "UNIMPLEMENTED",
    "INTERNAL", "UNAVAILABLE", "DATA_LOSS", "UNAUTHENTICATED",
]


class _Code:
    def __init__(self, value: int) -> None:
        self.value = value
        self.name = (
            _CODE_NAMES[value] if 0 <= value < len(_CODE_NAMES) else f"CODE_{value}"
        )


class NativeRpcError(Exception):
    """Duck-typed like grpc.RpcError (code().name / details())."""

    def __init__(self, code: int, message: str) -> None:
        super().__init__(f"{_Code(code).name}: {message}")
        self._code = _Code(code)
        self._message = message

    def code(self) -> _Code:
        return self._code

    def details(self) -> str:
        return self._message


def load_module():
    import importlib
    import sys
    from pathlib import Path

    ops_dir = str(Path(__file__).resolve().parent.parent / "ops")
    if ops_dir not in sys.path:
        sys.path.insert(0, ops_dir)
    return importlib.import_module("_h2grpc")


class NativeWireClient:
    """Batch unary invoker over one backend target."""

    def __init__(self, target: str, connections: int = 8, authority: str = "",
                 max_inflight: int = 512, max_resp_bytes: int = 0) -> None:
        mod = load_module()
        self.target = target
        self._cli = mod.Client(target, connections=connections,
                               authority=authority or "localhost",
                               max_inflight=max_inflight,
                               max_resp_bytes=max_resp_bytes)

    def invoke_batch(
        self,
        paths: Sequence[str],
        payloads: Sequence[bytes],
        timeout_s: float,
        metadata: Optional[Sequence[Sequence[Tuple[str, str]]]] = None,
    ) -> List[object]:
        """Returns a list of bytes (ok) or NativeRpcError per slot."""
        res = self._cli.invoke_batch(
            list(paths), list(payloads), timeout_s, list(metadata or [])
        )
        out: List[object] = []
        for status, payload, message in res:
            if status == 0:
                out.append(payload)
            else:
                out.append(NativeRpcError(status, message))
        return out

    def invoke_stream_batch(
        self,
        paths: Sequence[str],
        payloads: Sequence[bytes],
        timeout_s: float,
        metadata: Optional[Sequence[Sequence[Tuple[str, str]]]] = None,
    ) -> List[object]:
        """Server-streaming batch: each slot resolves to a list of wire
        chunks — zero-copy memoryview slices of one blob per stream — or
        NativeRpcError."""
        res = self._cli.invoke_stream_batch(
            list(paths), list(payloads), timeout_s, list(metadata or [])
        )
        out: List[object] = []
        for status, blob, lens, message in res:
            if status == 0:
                mv = memoryview(blob)
                chunks = []
                off = 0
                for ln in lens:
                    chunks.append(mv[off : off + ln])
                    off += ln
                out.append(chunks)
            else:
                out.append(NativeRpcError(status, message))
        return out

    def healthy(self) -> bool:
        return self._cli.healthy()

    def close(self) -> None:
        self._cli.close()
