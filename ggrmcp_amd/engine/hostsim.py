"""Python wrapper for the single-lane CPU build of the gfx950 kernels.

Mirrors ``GpuEngine.encode_batch``/``decode_batch`` (batch.py) exactly —
same compiled tables, same offset/arena layout — so the CPU test tier can
differential-test the kernel LOGIC against the protojson oracle on every
run, without a GPU (tests/test_hostsim.py).  See ops/csrc/host_shim.h for
why WAVE=1 execution is faithful.
"""

from __future__ import annotations

import importlib
import sys
from pathlib import Path
from typing import Any, Dict, List, Optional, Sequence, Tuple

import numpy as np

from .batch import DECODE_DTYPE, E_OK, SLOT_DTYPE, _offsets
from .tables import compile_tables


def load_module():
    ops_dir = str(Path(__file__).resolve().parent.parent / "ops")
    if ops_dir not in sys.path:
        sys.path.insert(0, ops_dir)
    return importlib.import_module("_hostsim")


class HostSimEngine:
    """CPU twin of GpuEngine (same tables, same buffer layout)."""

    def __init__(self, tools: Dict[str, Any]) -> None:
        mod = load_module()
        self.tables = compile_tables(tools)
        t = self.tables
        self._eng = mod.HostEngine(
            t.msg_table, t.field_table, t.enum_table, t.enum_values,
            t.tool_table, t.name_blob, t.n_msgs, t.n_tools,
        )

    def encode_batch(
        self,
        payloads: Sequence[bytes],
        mode: int,
        msg_indices: Optional[Sequence[int]] = None,
        enforce: bool = True,
    ) -> Tuple[np.ndarray, List[Optional[bytes]]]:
        data = b"".join(payloads)
        in_off = _offsets([len(p) for p in payloads])
        pb_off = _offsets([len(p) + len(p) // 4 for p in payloads], pad=192, align=16)
        msg_idx = (
            np.asarray(msg_indices, dtype=np.int32) if msg_indices is not None else None
        )
        raw, pb = self._eng.encode(
            data, in_off, pb_off, msg_idx, mode, 10, 1024, 1 << 20,
            1 if enforce else 0,
        )
        results = np.frombuffer(raw.tobytes(), dtype=SLOT_DTYPE)
        out: List[Optional[bytes]] = []
        for r in results:
            if r["status"] == E_OK:
                out.append(pb[r["pb_off"] : r["pb_off"] + r["pb_len"]])
            else:
                out.append(None)
        return results, out

    def decode_batch(
        self,
        payloads: Sequence[Optional[bytes]],
        msg_indices: Sequence[int],
        mode: int,
        skip: Optional[Sequence[bool]] = None,
    ) -> Tuple[np.ndarray, List[Optional[bytes]]]:
        skips = [1 if (p is None or (skip is not None and skip[i])) else 0
                 for i, p in enumerate(payloads)]
        safe = [p if p is not None else b"" for p in payloads]
        data = b"".join(safe)
        lens = [len(p) for p in safe]
        resp_off = _offsets(lens)
        scratch_off = _offsets([n * 8 + 1024 for n in lens], align=16)
        final_off = _offsets([n * 16 + 2048 for n in lens], align=16)
        raw, fin = self._eng.decode(
            data, resp_off, scratch_off, final_off,
            np.asarray(msg_indices, dtype=np.int32),
            np.asarray(skips, dtype=np.int32), mode,
        )
        results = np.frombuffer(raw.tobytes(), dtype=DECODE_DTYPE)
        out: List[Optional[bytes]] = []
        for i, r in enumerate(results):
            if skips[i] or r["status"] != E_OK:
                out.append(None)
            else:
                out.append(fin[r["out_off"] : r["out_off"] + r["out_len"]])
        return results, out
