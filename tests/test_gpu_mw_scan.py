"""Multi-wave speculative structural scan (wg encode phase A')
differential tests.

GGRMCP_MW_SCAN=1 routes big arguments objects through the parallel
window-index path (json2pb.hip phase A'); the produced items are spans
into the same source, so the wire must be BYTE-identical to both the
serial wg scanner and the classic per-wave kernel.  Any anomaly must
fall back to classic (identical error formats)."""

import json
import os
import random

import pytest

pytestmark = pytest.mark.gpu

from test_gpu_wg_decode import _body, _first_diff, _shapes, env  # noqa: F401,E402


def _enc(pipeline, bodies, mw, wg_min="4096"):
    os.environ["GGRMCP_WG_ENC_MIN"] = wg_min
    os.environ["GGRMCP_MW_SCAN"] = "1" if mw else "0"
    try:
        return pipeline.engine.encode_batch(bodies, mode=0)
    finally:
        os.environ.pop("GGRMCP_WG_ENC_MIN", None)
        os.environ.pop("GGRMCP_MW_SCAN", None)


def _assert_equal(bodies, a, b):
    enc_a, pbs_a = a
    enc_b, pbs_b = b
    for i in range(len(bodies)):
        assert enc_a[i]["status"] == enc_b[i]["status"], (
            i, enc_a[i]["status"], enc_b[i]["status"])
        assert enc_a[i]["tool_idx"] == enc_b[i]["tool_idx"], i
        assert enc_a[i]["err_pos"] == enc_b[i]["err_pos"], i
        assert enc_a[i]["aux"] == enc_b[i]["aux"], i
        assert enc_a[i]["id_len"] == enc_b[i]["id_len"], i
        assert pbs_a[i] == pbs_b[i], (
            f"slot {i} wire diverged: {len(pbs_a[i] or b'')}B vs "
            f"{len(pbs_b[i] or b'')}B")


def test_mw_scan_matches_serial_wg(env):  # noqa: F811
    pipeline, d = env
    bodies = [_body(a, i + 1) for i, a in enumerate(_shapes())]
    serial = _enc(pipeline, bodies, mw=False)
    mw = _enc(pipeline, bodies, mw=True)
    _assert_equal(bodies, serial, mw)


def test_mw_scan_matches_classic(env):  # noqa: F811
    pipeline, d = env
    bodies = [_body(a, i + 1) for i, a in enumerate(_shapes())]
    classic = _enc(pipeline, bodies, mw=False, wg_min="1000000000")
    mw = _enc(pipeline, bodies, mw=True)
    _assert_equal(bodies, classic, mw)


def test_mw_scan_fuzz(env):  # noqa: F811
    from ggrmcp_amd.utils.synthetic import wide_payload

    pipeline, d = env
    rng = random.Random(321)
    bodies = [
        _body(wide_payload(rng, target_bytes=rng.choice(
            [9000, 17000, 33000, 64000])), i + 1)
        for i in range(48)
    ]
    serial = _enc(pipeline, bodies, mw=False, wg_min="2048")
    mw = _enc(pipeline, bodies, mw=True, wg_min="2048")
    _assert_equal(bodies, serial, mw)


def test_mw_scan_anomalies_fall_back(env):  # noqa: F811
    """Unknown fields, duplicate members, escaped keys, deep nesting,
    truncated bodies: mw must land on the same (classic) output."""
    pipeline, d = env
    pad = {"attrs": {f"p{j}": "x" * 900 for j in range(24)}}
    bodies = [_body({"nosuchfield": 1, **pad}, 1)]
    dup_pad = ",".join(f'"z{j}": "' + "y" * 900 + '"' for j in range(12))
    bodies.append(
        ('{"jsonrpc":"2.0","id":2,"method":"tools/call","params":'
         '{"name":"bench_echoservice_echo","arguments":{'
         '"f02Int32":1,"f02Int32":2,"attrs":{' + dup_pad + "}}}}").encode())
    esc_pad = ",".join(f'"w{j}": "' + "v" * 900 + '"' for j in range(12))
    bodies.append(
        ('{"jsonrpc":"2.0","id":3,"method":"tools/call","params":'
         '{"name":"bench_echoservice_echo","arguments":{'
         '"f01Str\\u0069ng":"esc-key",' + esc_pad + "}}}").encode())
    deep = '{"child": ' * 40 + '{"name": "x"}' + "}" * 40
    bodies.append(
        ('{"jsonrpc":"2.0","id":4,"method":"tools/call","params":'
         '{"name":"bench_echoservice_echo","arguments":{'
         '"f40Node": ' + deep + ', "attrs":{' + esc_pad + "}}}}").encode())
    trunc = _body({"attrs": {f"q{j}": "t" * 900 for j in range(12)}}, 5)
    bodies.append(trunc[: len(trunc) - 7])
    serial = _enc(pipeline, bodies, mw=False, wg_min="1024")
    mw = _enc(pipeline, bodies, mw=True, wg_min="1024")
    _assert_equal(bodies, serial, mw)


def test_mw_scan_end_to_end_oracle(env):  # noqa: F811
    pipeline, d = env
    shapes = _shapes()
    bodies = [_body(a, i + 1) for i, a in enumerate(shapes)]
    os.environ["GGRMCP_MW_SCAN"] = "1"
    os.environ["GGRMCP_WG_ENC_MIN"] = "4096"
    try:
        out = pipeline.process_batch(bodies, timeout_s=30.0)
    finally:
        os.environ.pop("GGRMCP_MW_SCAN", None)
        os.environ.pop("GGRMCP_WG_ENC_MIN", None)
    mi = d.tools["bench_echoservice_echo"]
    for i, (args, raw) in enumerate(zip(shapes, out)):
        resp = json.loads(raw)
        assert resp["result"]["isError"] is False, resp
        inner = json.loads(resp["result"]["content"][0]["text"])
        wire = pipeline.cpu.json_to_pb(mi.input_descriptor, json.dumps(args))
        oracle = json.loads(pipeline.cpu.pb_to_json(mi.output_descriptor, wire))
        assert inner == oracle, f"slot {i}: {_first_diff(inner, oracle)}"


def test_mw_scan_adversarial_strings(env):  # noqa: F811
    """The speculative scanner's hard cases: structural characters INSIDE
    strings, backslash runs (escape state crossing 256 B window
    boundaries), and tokens straddling window edges.  Wire must stay
    byte-identical to the serial scanner."""
    rng = random.Random(999)
    bodies = []
    rid = 1

    def mk(args):
        nonlocal rid
        bodies.append(_body(args, rid))
        rid += 1

    # strings stuffed with JSON structural chars (validation caps strings
    # at 1024 chars, so spread across many map values)
    evil = '{"a":[1,2],\\\\}' + "{[,]}" * 60
    mk({"attrs": {f"k{j}": (evil * 3)[:900] for j in range(24)}})
    # long backslash runs with varying parity near 256-byte boundaries
    for parity in (1, 2, 3):
        v = ("x" * 251 + "\\" * parity + '"inner"').replace('"inner"', "")
        # json.dumps escapes backslashes: each '\\' doubles on the wire,
        # shifting window alignment per value
        mk({"attrs": {f"b{j}": v + "y" * (j % 7) for j in range(20)}})
    # quotes adjacent to escapes: \" \\" \\\" patterns
    q = 'a\\"b' + "\\\\" + '\\"' + "c"
    mk({"attrs": {f"q{j}": (q * 40)[:800] for j in range(20)}})
    # member keys landing near window boundaries: tune value sizes so
    # keys fall at 256k +- 2 offsets
    for shift in range(-2, 3):
        mk({"attrs": {f"w{j:03d}": "z" * (249 + shift) for j in range(40)}})
    # deep nesting inside a big payload (within limits)
    nest = {"name": "leaf"}
    for _ in range(8):
        nest = {"child": nest, "name": "n" * 50}
    mk({"f40Node": nest,
        "attrs": {f"n{j}": "m" * 700 for j in range(16)}})
    # unicode multibyte spanning boundaries
    mk({"attrs": {f"u{j}": ("é中\U0001f600" * 80)[:300] for j in range(30)}})

    pipeline, d = env
    serial = _enc(pipeline, bodies, mw=False, wg_min="2048")
    mw = _enc(pipeline, bodies, mw=True, wg_min="2048")
    _assert_equal(bodies, serial, mw)
