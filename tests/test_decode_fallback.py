"""Decode-stage fallback semantics (CPU): a response that was DELIVERED but
rejected by the GPU decode stage must be transcoded from the received bytes
— never re-invoked (duplicate side effects on non-idempotent RPCs; VERDICT
r1 item 2 / ADVICE batch.py:680).  The GPU-marked twin
(test_gpu_pipeline.py::test_badutf8_response_single_invoke) runs the real
kernel + native transport path end-to-end."""

import json

import numpy as np
import pytest

from examples.protos import ALL_FDPS
from ggrmcp_amd.descriptors.loader import build_pool, extract_method_infos
from ggrmcp_amd.engine.batch import (
    DECODE_DTYPE,
    E_OK,
    E_OVERFLOW,
    E_UNSUPPORTED,
    EngineStats,
    GpuPipeline,
    SLOT_DTYPE,
)
from ggrmcp_amd.engine.cpu_ref import CpuTranscoder


class CountingDiscoverer:
    def __init__(self):
        self.invokes = 0

    def invoke_method_by_tool(self, tool, args_json, hdr, timeout_s):
        self.invokes += 1
        return "{}"


class _Eng:
    def __init__(self):
        self.stats = EngineStats()


def _mk_pipeline():
    """GpuPipeline shell with just the fields _host_slot touches — no GPU."""
    pool = build_pool(ALL_FDPS)
    infos = extract_method_infos(ALL_FDPS, pool, compat_names=False)
    tools = {mi.tool_name(): mi for mi in infos}
    p = GpuPipeline.__new__(GpuPipeline)
    p.cpu = CpuTranscoder()
    p.discoverer = CountingDiscoverer()
    p._mi_by_idx = [tools["hello_helloservice_sayhello"]]
    return p


def _enc(status=E_OK, tool_idx=0):
    r = np.zeros(1, dtype=SLOT_DTYPE)[0]
    r["status"] = status
    r["tool_idx"] = tool_idx
    return r


def _dec(status):
    r = np.zeros(1, dtype=DECODE_DTYPE)[0]
    r["status"] = status
    return r


BODY = json.dumps(
    {"jsonrpc": "2.0", "id": 7, "method": "tools/call",
     "params": {"name": "hello_helloservice_sayhello", "arguments": {}}}
).encode()

# hello.HelloResponse{message: "hi"} wire bytes
WIRE_OK = b"\x0a\x02hi"
# message field containing invalid UTF-8 (protojson rejects it)
WIRE_BADUTF8 = b"\x0a\x03a\xff\xfe"


def test_delivered_wire_transcoded_not_reinvoked():
    p = _mk_pipeline()
    eng = _Eng()
    out = p._host_slot(eng, BODY, _enc(), _dec(E_UNSUPPORTED), WIRE_OK,
                       None, None, 5.0)
    resp = json.loads(out)
    assert resp["id"] == 7
    assert resp["result"]["isError"] is False
    assert json.loads(resp["result"]["content"][0]["text"]) == {"message": "hi"}
    assert p.discoverer.invokes == 0, "decode fallback must NOT re-invoke"
    assert eng.stats.host_fallbacks == 1


def test_undecodable_wire_becomes_error_without_reinvoke():
    """Invalid UTF-8 in the response: even the CPU oracle rejects it, so the
    slot yields an internal error — still with exactly zero extra invokes."""
    p = _mk_pipeline()
    eng = _Eng()
    out = p._host_slot(eng, BODY, _enc(), _dec(E_UNSUPPORTED), WIRE_BADUTF8,
                       None, None, 5.0)
    resp = json.loads(out)
    assert resp["id"] == 7
    assert "error" in resp
    assert p.discoverer.invokes == 0


def test_encode_side_fallback_invokes_exactly_once():
    """E_UNSUPPORTED at ENCODE means nothing was sent yet: the CPU path owns
    the one and only invoke."""
    p = _mk_pipeline()
    eng = _Eng()
    out = p._host_slot(eng, BODY, _enc(status=E_UNSUPPORTED), None, None,
                       None, None, 5.0)
    resp = json.loads(out)
    assert resp["result"]["isError"] is False
    assert p.discoverer.invokes == 1


def test_decode_overflow_uses_wire():
    p = _mk_pipeline()
    eng = _Eng()
    out = p._host_slot(eng, BODY, _enc(), _dec(E_OVERFLOW), WIRE_OK,
                       None, None, 5.0)
    resp = json.loads(out)
    assert resp["result"]["isError"] is False
    assert p.discoverer.invokes == 0
