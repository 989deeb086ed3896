"""Chunked copy/compute pipeline differential tests (engine.cpp
run_encode_chunked / chunked decode H2D).

Large batches split into byte-balanced item chunks so chunk c+1's H2D and
chunk c-1's D2H overlap chunk c's kernels (stream2_).  A sub-range launch
is shifted pointers + a smaller n — these tests force the chunked path
over the same payloads as the single-shot path (GGRMCP_PIPE_MIN /
GGRMCP_PIPE_CHUNKS) and require byte-identical wire in both directions,
for both the classic and workgroup-cooperative kernels."""

import json
import os
import random

import pytest

pytestmark = pytest.mark.gpu

from test_gpu_wg_decode import _body, _first_diff, _shapes, env  # noqa: F401,E402


def _mixed_bodies(n=24, seed=42):
    from ggrmcp_amd.utils.synthetic import wide_payload

    rng = random.Random(seed)
    bodies = []
    for i in range(n):
        kind = i % 3
        if kind == 0:  # small serving-shaped
            args = {"f01String": "hello" * rng.randint(1, 8), "f02Int32": i}
        elif kind == 1:  # medium
            args = wide_payload(rng, target_bytes=9000)
        else:  # large (wg-eligible)
            args = wide_payload(rng, target_bytes=48000)
        bodies.append(_body(args, i + 1))
    return bodies


def _enc(pipeline, bodies, chunks, pipe_min, wg_min=None):
    os.environ["GGRMCP_PIPE_CHUNKS"] = str(chunks)
    os.environ["GGRMCP_PIPE_MIN"] = str(pipe_min)
    if wg_min is not None:
        os.environ["GGRMCP_WG_ENC_MIN"] = str(wg_min)
    try:
        return pipeline.engine.encode_batch(bodies, mode=0)
    finally:
        for k in ("GGRMCP_PIPE_CHUNKS", "GGRMCP_PIPE_MIN",
                  "GGRMCP_WG_ENC_MIN"):
            os.environ.pop(k, None)


def _assert_enc_equal(bodies, a, b):
    enc_a, pbs_a = a
    enc_b, pbs_b = b
    for i in range(len(bodies)):
        assert enc_a[i]["status"] == enc_b[i]["status"], i
        assert enc_a[i]["tool_idx"] == enc_b[i]["tool_idx"], i
        assert enc_a[i]["id_len"] == enc_b[i]["id_len"], i
        assert pbs_a[i] == pbs_b[i], (
            f"slot {i} wire diverged: {len(pbs_a[i] or b'')}B vs "
            f"{len(pbs_b[i] or b'')}B")


def test_chunked_encode_matches_single_shot(env):  # noqa: F811
    pipeline, d = env
    bodies = _mixed_bodies()
    single = _enc(pipeline, bodies, chunks=1, pipe_min=1)
    for c in (2, 3, 4, 8):
        chunked = _enc(pipeline, bodies, chunks=c, pipe_min=1)
        _assert_enc_equal(bodies, single, chunked)


def test_chunked_encode_with_wg_routing(env):  # noqa: F811
    """Chunk cuts must not disturb the wg sub-range launches."""
    pipeline, d = env
    bodies = _mixed_bodies(n=16, seed=7)
    single = _enc(pipeline, bodies, chunks=1, pipe_min=1, wg_min=2048)
    chunked = _enc(pipeline, bodies, chunks=4, pipe_min=1, wg_min=2048)
    _assert_enc_equal(bodies, single, chunked)


def test_chunked_end_to_end_oracle(env):  # noqa: F811
    """Forced-chunked encode+decode through the full pipeline stays
    protojson-exact (decode chunking needs mode-0 has_skip, which the
    pipeline always sets)."""
    pipeline, d = env
    shapes = _shapes()
    bodies = [_body(a, i + 1) for i, a in enumerate(shapes)]
    os.environ["GGRMCP_PIPE_CHUNKS"] = "4"
    os.environ["GGRMCP_PIPE_MIN"] = "1"
    try:
        out = pipeline.process_batch(bodies, timeout_s=30.0)
    finally:
        os.environ.pop("GGRMCP_PIPE_CHUNKS", None)
        os.environ.pop("GGRMCP_PIPE_MIN", None)
    mi = d.tools["bench_echoservice_echo"]
    for i, (args, raw) in enumerate(zip(shapes, out)):
        resp = json.loads(raw)
        assert resp["result"]["isError"] is False, resp
        inner = json.loads(resp["result"]["content"][0]["text"])
        wire = pipeline.cpu.json_to_pb(mi.input_descriptor, json.dumps(args))
        oracle = json.loads(pipeline.cpu.pb_to_json(mi.output_descriptor, wire))
        assert inner == oracle, f"slot {i}: {_first_diff(inner, oracle)}"


def test_chunked_end_to_end_errors_and_default_off(env):  # noqa: F811
    """Error slots keep their envelopes under chunking; a small batch
    below GGRMCP_PIPE_MIN takes the single-shot path unchanged."""
    pipeline, d = env
    bodies = [
        _body({"f01String": "ok", "f02Int32": 1}, 1),
        b'{"jsonrpc":"2.0","id":2,"method":"tools/call","params":'
        b'{"name":"bench_echoservice_echo","arguments":{"nosuch":1}}}',
        _body({"f01String": "ok2"}, 3),
    ]
    os.environ["GGRMCP_PIPE_CHUNKS"] = "3"
    os.environ["GGRMCP_PIPE_MIN"] = "1"
    try:
        chunked = pipeline.process_batch(bodies, timeout_s=30.0)
    finally:
        os.environ.pop("GGRMCP_PIPE_CHUNKS", None)
        os.environ.pop("GGRMCP_PIPE_MIN", None)
    plain = pipeline.process_batch(bodies, timeout_s=30.0)
    for i, (a, b) in enumerate(zip(chunked, plain)):
        assert json.loads(a) == json.loads(b), i


def test_staging_pool_equivalence(env):  # noqa: F811
    """GGRMCP_STAGE_THREADS fans the span staging memcpys across a helper
    pool (default off — measured neutral).  Forced on, the full pipeline
    must produce identical responses."""
    pipeline, d = env
    bodies = _mixed_bodies(n=24, seed=77)
    plain = pipeline.process_batch(bodies, timeout_s=30.0)
    os.environ["GGRMCP_STAGE_THREADS"] = "3"
    os.environ["GGRMCP_STAGE_MIN"] = "1"
    try:
        pooled = pipeline.process_batch(bodies, timeout_s=30.0)
    finally:
        os.environ.pop("GGRMCP_STAGE_THREADS", None)
        os.environ.pop("GGRMCP_STAGE_MIN", None)
    for i, (a, b) in enumerate(zip(plain, pooled)):
        assert json.loads(a) == json.loads(b), i
