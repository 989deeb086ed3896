"""Workgroup-cooperative decode (k_pb2json_wg) differential tests.

Large responses route one-workgroup-per-request (common.h WG_DEC_*); the
cooperative path must be byte-identical to the classic per-wave kernel and
structurally identical to the protojson oracle.  GGRMCP_WG_DEC_MIN lets
the test force BOTH paths over the same payloads in one process."""

import json
import os
import random

import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def env():
    from google.protobuf import descriptor_pb2

    from examples.protos import ALL_FDPS
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer
    from ggrmcp_amd.backend.native_invoker import NativeWireClient, load_module
    from ggrmcp_amd.config import Config
    from ggrmcp_amd.engine.batch import GpuPipeline
    from ggrmcp_amd.utils.synthetic import synthetic_fdp

    mod = load_module()
    srv = mod.Server("127.0.0.1:0")
    srv.add_route("/bench.EchoService/Echo", "echo")
    bound = srv.start()
    cfg = Config.default()
    host, _, port = bound.rpartition(":")
    cfg.grpc.host, cfg.grpc.port = host, int(port)
    # 64 KB payloads need the big arenas (bench.py wide64 settings)
    cfg.gpu.pinned_pool_bytes = 1 * 1024 * 1024 * 1024
    cfg.gpu.device_pool_bytes = 4 * 1024 * 1024 * 1024
    cfg.gpu.streams = 1
    d = ServiceDiscoverer(cfg)
    fdset = descriptor_pb2.FileDescriptorSet()
    fdset.file.extend(ALL_FDPS + [synthetic_fdp()])
    d.load_descriptor_blob(fdset.SerializeToString())
    d.connections[0].connect(timeout_s=15)
    wire = NativeWireClient(bound, connections=2)
    pipeline = GpuPipeline(d, cfg, device=0, wire_clients=[wire])
    yield pipeline, d
    os.environ.pop("GGRMCP_WG_DEC_MIN", None)
    wire.close()
    d.close()
    srv.stop()


def _first_diff(a, b, path="$"):
    if isinstance(a, dict) and isinstance(b, dict):
        for k in sorted(set(a) | set(b)):
            if a.get(k) != b.get(k):
                return _first_diff(a.get(k), b.get(k), f"{path}.{k}")
    if isinstance(a, list) and isinstance(b, list):
        if len(a) != len(b):
            return f"{path}: len {len(a)} vs {len(b)}"
        for j, (x, y) in enumerate(zip(a, b)):
            if x != y:
                return _first_diff(x, y, f"{path}[{j}]")
    return f"{path}: {a!r} != {b!r}"


def _body(args, rid):
    return json.dumps(
        {"jsonrpc": "2.0", "id": rid, "method": "tools/call",
         "params": {"name": "bench_echoservice_echo", "arguments": args}}
    ).encode()


def _shapes():
    """Payload shapes stressing the item scanner/joiner."""
    from ggrmcp_amd.utils.synthetic import wide_payload

    rng = random.Random(7)
    shapes = []
    # config-3 standard: 64-field nested proto, ~64 KB (strings stay under
    # the MCP 1024-char validation cap — that cap applies on ENCODE)
    shapes.append(wide_payload(rng, target_bytes=64 * 1024))
    # ONE top-level item carrying the whole payload (a single map run)
    shapes.append({"attrs": {f"p{j}": "x" * 1000 for j in range(40)}})
    # escapes + unicode spread across items
    shapes.append({
        # x20: json.dumps ascii-escapes the unicode, and the \uXXXX form is
        # what the 1024-char string limit sees on the wire
        "f01String": ('he said "hi"\n\t\\' + "é中\U0001f600") * 20,
        "f02Int32": -7,
        "f05Bool": True,
        "items": [{"key": f"i{j}", "value": str(j), "weight": j / 3}
                  for j in range(400)],
        "attrs": {f"q{j}": "y" * 500 for j in range(20)},
    })
    # many items: scalars + repeated + map + nested around the threshold
    shapes.append(wide_payload(rng, target_bytes=20 * 1024))
    # default-valued singulars between real fields (empty items -> no
    # stray commas)
    shapes.append({
        # field-number order (the kernels' single-pass subset expects
        # ascending wire fields; JSON order drives the encoder's emission)
        "f01String": "a" * 1000,
        "f02Int32": 0,          # proto3 default: omitted from output
        "f03Int64": "0",
        "f04Double": 1.5,
        "f05Bool": False,
        "attrs": {f"z{j}": "w" * 900 for j in range(16)},
    })
    # doubles / int64 precision paths at volume (repeated Inner)
    shapes.append({
        "f01String": "p" * 1000,
        "items": [{"key": f"d{j}",
                   "value": str(rng.randint(-(2**40), 2**40)),
                   "weight": rng.random() * 10 ** rng.randint(-12, 12)}
                  for j in range(700)],
        "attrs": {f"k{j}": "v" * rng.randint(0, 40) for j in range(120)},
    })
    return shapes


def test_wg_decode_matches_classic_bytes(env):
    """Classic per-wave vs cooperative outputs must be BYTE-identical."""
    pipeline, d = env
    bodies = [_body(a, i + 1) for i, a in enumerate(_shapes())]

    os.environ["GGRMCP_WG_DEC_MIN"] = "1000000000"  # force classic
    classic = pipeline.process_batch(bodies, timeout_s=30.0)
    os.environ["GGRMCP_WG_DEC_MIN"] = "8192"        # force cooperative
    coop = pipeline.process_batch(bodies, timeout_s=30.0)
    os.environ.pop("GGRMCP_WG_DEC_MIN", None)

    for i, (a, b) in enumerate(zip(classic, coop)):
        assert a == b, f"slot {i} diverged:\n{a[:400]}\nvs\n{b[:400]}"
        resp = json.loads(b)
        assert resp["result"]["isError"] is False, resp


def test_wg_decode_matches_protojson(env):
    """Cooperative output vs the protojson oracle (structural equality:
    the echo backend returns the request message, so the oracle is
    protojson(json_to_pb(args)))."""
    pipeline, d = env
    shapes = _shapes()
    bodies = [_body(a, i + 1) for i, a in enumerate(shapes)]
    fallbacks_before = pipeline.engine.stats.host_fallbacks
    out = pipeline.process_batch(bodies, timeout_s=30.0)
    mi = d.tools["bench_echoservice_echo"]
    for i, (args, raw) in enumerate(zip(shapes, out)):
        resp = json.loads(raw)
        assert resp["id"] == i + 1
        assert resp["result"]["isError"] is False, resp
        inner = json.loads(resp["result"]["content"][0]["text"])
        wire = pipeline.cpu.json_to_pb(mi.input_descriptor, json.dumps(args))
        oracle = json.loads(pipeline.cpu.pb_to_json(mi.output_descriptor, wire))
        assert inner == oracle, f"slot {i}: {_first_diff(inner, oracle)}"
    assert pipeline.engine.stats.host_fallbacks == fallbacks_before, (
        "every shape must complete on the GPU path")


def test_wg_decode_fuzz_vs_classic(env):
    """Randomized wide payloads through both kernels, byte-compared."""
    from ggrmcp_amd.utils.synthetic import wide_payload

    pipeline, d = env
    rng = random.Random(99)
    bodies = [
        _body(wide_payload(rng, target_bytes=rng.choice(
            [12000, 18000, 33000, 64000])), i)
        for i in range(48)
    ]
    os.environ["GGRMCP_WG_DEC_MIN"] = "1000000000"
    classic = pipeline.process_batch(bodies, timeout_s=30.0)
    os.environ["GGRMCP_WG_DEC_MIN"] = "4096"
    coop = pipeline.process_batch(bodies, timeout_s=30.0)
    os.environ.pop("GGRMCP_WG_DEC_MIN", None)
    for i, (a, b) in enumerate(zip(classic, coop)):
        assert a == b, f"slot {i} diverged"
