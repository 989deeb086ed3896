"""Property-based differential fuzz: kernel logic (host build) vs protojson.

Hypothesis generates random Wide64/Node payloads (every scalar kind, nested
messages, repeated, map, enum, unicode strings, boundary numbers); every
example must encode and decode identically to google.protobuf.json_format.
Complements the fixed cases in test_hostsim.py.
"""

import json
import math
import string

import pytest
from hypothesis import given, settings, strategies as st

from examples.protos import ALL_FDPS
from ggrmcp_amd.descriptors.loader import build_pool, extract_method_infos
from ggrmcp_amd.engine.cpu_ref import CpuTranscoder
from ggrmcp_amd.engine.hostsim import HostSimEngine
from ggrmcp_amd.utils.synthetic import synthetic_fdp
from google.protobuf import json_format

_fdps = ALL_FDPS + [synthetic_fdp()]
_pool = build_pool(_fdps)
_infos = {m.tool_name(): m for m in extract_method_infos(_fdps, _pool, compat_names=False)}
_engine = HostSimEngine(_infos)
_cpu = CpuTranscoder()


def _approx(a, b, rel=1e-9):
    if isinstance(a, dict) and isinstance(b, dict):
        return a.keys() == b.keys() and all(_approx(a[k], b[k], rel) for k in a)
    if isinstance(a, list) and isinstance(b, list):
        return len(a) == len(b) and all(_approx(x, y, rel) for x, y in zip(a, b))
    if isinstance(a, float) or isinstance(b, float):
        if isinstance(a, str) or isinstance(b, str):
            return str(a) == str(b)
        return math.isclose(float(a), float(b), rel_tol=rel, abs_tol=1e-300)
    return a == b


# text without lone surrogates (protojson rejects them too)
_text = st.text(
    alphabet=st.characters(blacklist_categories=("Cs",), max_codepoint=0x10FFFF),
    max_size=40,
)
_ascii = st.text(alphabet=string.ascii_letters + string.digits + "_-. \t\n\"\\/", max_size=48)

_i32 = st.integers(min_value=-(2**31), max_value=2**31 - 1)
_i64s = st.integers(min_value=-(2**63), max_value=2**63 - 1).map(str)
_u32 = st.integers(min_value=0, max_value=2**32 - 1)
_u64s = st.integers(min_value=0, max_value=2**64 - 1).map(str)
_dbl = st.one_of(
    st.floats(allow_nan=False, allow_infinity=False, width=64),
    st.sampled_from([0.0, -0.0, 1e-300, 1e300, 2.2250738585072014e-308]),
    # dyadic rationals: exact in fp64, tie-prone decimal digit strings
    st.builds(lambda m, k: float(m) / (1 << k),
              st.integers(min_value=1, max_value=2**40 - 1),
              st.integers(min_value=0, max_value=40)),
)
# dyadic rationals m*2^-k land exactly between short decimal strings and
# hammer the formatter's round-half-to-even tie-break (fuzz caught fp32
# 1048576.25 printing ".3" where Ryu/protojson print ".2")
_tie_flt = st.builds(
    lambda m, k: float(m) / (1 << k),
    st.integers(min_value=1, max_value=2**24 - 1),
    st.integers(min_value=0, max_value=24),
)
# m < 2^24 always fits fp32's 24-bit significand, so these are exact
_flt = st.one_of(
    st.floats(allow_nan=False, allow_infinity=False, width=32),
    _tie_flt,
)


def wide_strategy():
    inner = st.fixed_dictionaries(
        {},
        optional={
            "key": _ascii,
            "value": _i64s,
            "weight": _dbl,
        },
    )
    return st.fixed_dictionaries(
        {},
        optional={
            "f01String": _text,
            "f02Int32": _i32,
            "f03Int64": _i64s,
            "f04Double": _dbl,
            "f05Bool": st.booleans(),
            "f06Uint32": _u32,
            "f07Uint64": _u64s,
            "f08Float": _flt,
            "f09Sint32": _i32,
            "f10Sint64": _i64s,
            "f11Fixed32": _u32,
            "f12Fixed64": _u64s,
            "f13Sfixed32": _i32,
            "f14Sfixed64": _i64s,
            "nested": inner,
            "items": st.lists(inner, max_size=4),
            "level": st.sampled_from(["LEVEL_UNSET", "LOW", "MID", "HIGH", 0, 1, 3]),
            "attrs": st.dictionaries(_ascii.filter(bool), _text, max_size=4),
        },
    )


def node_strategy():
    return st.recursive(
        st.fixed_dictionaries({}, optional={"value": _text}),
        lambda children: st.fixed_dictionaries(
            {}, optional={"value": _text, "children": st.lists(children, max_size=3)}
        ),
        max_leaves=8,
    )


@settings(max_examples=120, deadline=None)
@given(payload=wide_strategy())
def test_fuzz_wide64_encode_decode(payload):
    _roundtrip("bench.Wide64", payload)


def test_wide64_float_boundary_regressions():
    """Pinned fuzz finds: top-of-range values where nextafter toward a
    finite constant (3.4e38f < FLT_MAX) stepped DOWNWARD, inverting the
    round-trip half-gap and rejecting the shortest digit form."""
    import struct as _struct

    fmax = _struct.unpack("<f", _struct.pack("<I", 0x7F7FFFFF))[0]  # FLT_MAX
    fsub = _struct.unpack("<f", _struct.pack("<I", 0x7F7FFFFE))[0]  # pred
    for v in (3.4028224522648084e+38, fmax, -fmax, fsub, -fsub,
              1.7976931348623157e+308, -1.7976931348623157e+308,
              1.0000000000000002e+308):
        _roundtrip("bench.Wide64", {"f08Float": min(max(v, -fmax), fmax)})
        _roundtrip("bench.Wide64", {"f04Double": v})


@settings(max_examples=60, deadline=None)
@given(node=node_strategy())
def test_fuzz_node_recursive(node):
    _roundtrip("complex.NodeRequest", {"root": node})


def _roundtrip(msg_name, payload):
    desc = _pool.FindMessageTypeByName(msg_name)
    text = json.dumps(payload, ensure_ascii=False)
    idx = _engine.tables.msg_index[msg_name]

    # encode: GPU-kernel wire must parse to the same message as the oracle
    enc, pbs = _engine.encode_batch([text.encode()], mode=1, msg_indices=[idx],
                                    enforce=False)
    try:
        oracle_wire = _cpu.json_to_pb(desc, text)
        oracle_ok = True
    except Exception:
        oracle_ok = False
    if not oracle_ok:
        # oracle rejects it; the kernel must reject too (any nonzero status)
        assert enc[0]["status"] != 0
        return
    if enc[0]["status"] == 6:  # E_UNSUPPORTED: declared GPU subset boundary
        # (e.g. escaped map keys) -> the pipeline host-transcodes, counted
        return
    assert enc[0]["status"] == 0, f"status {enc[0]['status']} aux={enc[0]['aux']} for {text!r}"
    g = json_format.MessageToDict(_cpu.pb_to_message(desc, pbs[0]))
    o = json_format.MessageToDict(_cpu.pb_to_message(desc, oracle_wire))
    assert _approx(g, o), f"\npayload: {text!r}\nsim:    {g}\noracle: {o}"

    # decode: oracle wire -> kernel JSON == oracle JSON
    dec, outs = _engine.decode_batch([oracle_wire], [idx], mode=1)
    if dec[0]["status"] == 6:  # declared subset boundary -> host fallback
        return
    assert dec[0]["status"] == 0, f"decode status {dec[0]['status']} for {text!r}"
    gj = json.loads(outs[0])
    oj = json.loads(_cpu.pb_to_json(desc, oracle_wire))
    assert _approx(gj, oj), f"\npayload: {text!r}\nsim:    {gj}\noracle: {oj}"


# ---- arbitrary JSON through google.protobuf.Struct --------------------------

from ggrmcp_amd.utils.protobuild import FileBuilder as _FB

_fb = _FB("t/struct.proto", "t")
_fb.add_dependency("google/protobuf/struct.proto")
_fb.message("Holder").field("data", 1, "message",
                            message="google.protobuf.Struct").done()
_fb.service("S").method("M", "Holder", "Holder").done()
_sfdp = _fb.build()
_spool = build_pool([_sfdp])
from ggrmcp_amd.descriptors.loader import extract_method_infos as _emi
_sengine = HostSimEngine({m.tool_name(): m
                          for m in _emi([_sfdp], _spool, compat_names=False)})
_sdesc = _spool.FindMessageTypeByName("t.Holder")

_json_value = st.recursive(
    st.one_of(st.none(), st.booleans(),
              st.floats(allow_nan=False, allow_infinity=False, width=64),
              _text),
    lambda v: st.one_of(st.lists(v, max_size=4),
                        st.dictionaries(_ascii, v, max_size=4)),
    max_leaves=12,
)


@settings(max_examples=100, deadline=None)
@given(value=st.dictionaries(_ascii, _json_value, max_size=5))
def test_fuzz_struct_value(value):
    payload = {"data": value}
    text = json.dumps(payload, ensure_ascii=False)
    idx = _sengine.tables.msg_index["t.Holder"]
    enc, pbs = _sengine.encode_batch([text.encode()], mode=1, msg_indices=[idx],
                                     enforce=False)
    try:
        oracle_wire = _cpu.json_to_pb(_sdesc, text)
    except Exception:
        assert enc[0]["status"] != 0
        return
    if enc[0]["status"] in (6, 7):
        # E_UNSUPPORTED / E_OVERFLOW: declared GPU-subset boundary (e.g.
        # pathological nesting where 3-byte len slots outgrow the pb cap);
        # the pipeline host-transcodes these, counted
        return
    assert enc[0]["status"] == 0, f"status {enc[0]['status']} for {text!r}"
    g = json_format.MessageToDict(_cpu.pb_to_message(_sdesc, pbs[0]))
    o = json_format.MessageToDict(_cpu.pb_to_message(_sdesc, oracle_wire))
    assert _approx(g, o), f"\npayload: {text!r}\nsim:    {g}\noracle: {o}"
    dec, outs = _sengine.decode_batch([oracle_wire], [idx], mode=1)
    if dec[0]["status"] == 6:
        return
    assert dec[0]["status"] == 0, f"decode status {dec[0]['status']} for {text!r}"
    gj = json.loads(outs[0])
    oj = json.loads(_cpu.pb_to_json(_sdesc, oracle_wire))
    assert _approx(gj, oj), f"\npayload: {text!r}\nsim:    {gj}\noracle: {oj}"


# ---- envelope mode: JSON-RPC validation parity ------------------------------

_env_engine = _engine  # same tool tables

_tool_names = st.sampled_from([
    "hello_helloservice_sayhello",        # valid unary
    "complex_nodeservice_streamnodes",    # valid server-streaming
    "no_such_tool",                       # -> E_METHOD_NOT_FOUND
])
_rpc_ids = st.one_of(st.integers(min_value=-(10**12), max_value=10**12),
                     st.text(alphabet=string.ascii_letters + "-", max_size=24),
                     st.none())
_methods = st.sampled_from(["tools/call", "tools/list", "initialize", "bogus/x"])


@settings(max_examples=150, deadline=None)
@given(rid=_rpc_ids, method=_methods, tool=_tool_names,
       args=st.dictionaries(st.sampled_from(["name"]), _text, max_size=1),
       jsonrpc=st.sampled_from(["2.0", "1.0", None]),
       drop_id=st.booleans())
def test_fuzz_envelope_statuses(rid, method, tool, args, jsonrpc, drop_id):
    body = {"method": method, "params": {"name": tool, "arguments": args}}
    if jsonrpc is not None:
        body["jsonrpc"] = jsonrpc
    if not drop_id:
        body["id"] = rid
    text = json.dumps(body, ensure_ascii=False).encode()
    enc, pbs = _env_engine.encode_batch([text], mode=0)
    status = int(enc[0]["status"])
    flags = int(enc[0]["flags"])
    if jsonrpc != "2.0":
        assert status == 2, (status, body)  # E_INVALID_REQUEST
        return
    if method != "tools/call":
        assert status == 8, (status, body)  # E_NOT_TOOLCALL
        return
    if tool == "no_such_tool":
        assert status == 3, (status, body)  # E_METHOD_NOT_FOUND
        return
    if tool == "complex_nodeservice_streamnodes":
        if args:
            # NodeRequest has no "name" field -> unknown field rejected
            # (protojson parity, builder.go strictness)
            assert status == 4, (status, body)
            return
        assert status == 0, (status, int(enc[0]["aux"]), body)
        assert flags & 2, flags  # SR_SERVER_STREAMING
    else:
        assert status == 0, (status, int(enc[0]["aux"]), body)
    if drop_id:
        assert flags & 1, flags  # SR_ID_IS_MISSING
    else:
        # the raw JSON id token is captured for the response envelope
        assert int(enc[0]["id_len"]) > 0, enc[0]


# ---- hostile input: arbitrary bytes must fail CLEANLY -----------------------

@settings(max_examples=300, deadline=None)
@given(data=st.binary(max_size=300))
def test_fuzz_hostile_envelope_bytes(data):
    """Random bytes through envelope mode: any E_* status is fine, crashes
    and out-of-range statuses are not."""
    enc, pbs = _engine.encode_batch([data], mode=0)
    assert 0 <= int(enc[0]["status"]) <= 8


@settings(max_examples=200, deadline=None)
@given(prefix=st.binary(max_size=120), cut=st.integers(min_value=0, max_value=400))
def test_fuzz_truncated_valid_envelope(prefix, cut):
    """A valid envelope truncated at any byte, with optional garbage glued
    on front, must fail cleanly (or parse if the cut lands at the end)."""
    body = json.dumps({"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                       "params": {"name": "hello_helloservice_sayhello",
                                  "arguments": {"name": "abc"}}}).encode()
    data = body[: min(cut, len(body))]
    enc, _ = _engine.encode_batch([data, prefix + data], mode=0)
    assert 0 <= int(enc[0]["status"]) <= 8
    assert 0 <= int(enc[1]["status"]) <= 8


@settings(max_examples=200, deadline=None)
@given(data=st.binary(max_size=200),
       msg=st.sampled_from(["bench.Wide64", "complex.NodeRequest",
                            "complex.Document", "complex.UserProfile"]))
def test_fuzz_hostile_wire_decode(data, msg):
    """Random bytes as protobuf wire through the decoder (several message
    shapes: wide scalars, recursion, maps+oneofs): clean status, and
    successful decodes must emit valid JSON (incl. valid UTF-8 — this
    fuzz caught the decoder forwarding invalid string-field bytes)."""
    idx = _engine.tables.msg_index[msg]
    dec, outs = _engine.decode_batch([data], [idx], mode=1)
    s = int(dec[0]["status"])
    assert 0 <= s <= 8
    if s == 0:
        json.loads(outs[0])  # must be well-formed
