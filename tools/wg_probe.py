#!/usr/bin/env python3
"""Minimal wg-decode probe: one payload through encode+invoke-less decode.

Usage: python tools/wg_probe.py <max_phase> [payload_kind]
Run each probe under `timeout` from the caller; prints PROBE-OK on success.
"""
import json
import os
import random
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

os.environ.setdefault("GGRMCP_WG_DEC_MIN", "64")
os.environ["GGRMCP_WG_PHASES"] = sys.argv[1] if len(sys.argv) > 1 else "3"
kind = sys.argv[2] if len(sys.argv) > 2 else "attrs"

from google.protobuf import descriptor_pb2

from ggrmcp_amd.backend.discovery import ServiceDiscoverer
from ggrmcp_amd.config import Config
from ggrmcp_amd.engine.batch import GpuEngine
from ggrmcp_amd.engine.cpu_ref import CpuTranscoder
from ggrmcp_amd.utils.synthetic import synthetic_fdp, wide_payload

cfg = Config.default()
d = ServiceDiscoverer(cfg)
fdset = descriptor_pb2.FileDescriptorSet()
fdset.file.extend([synthetic_fdp()])
d.load_descriptor_blob(fdset.SerializeToString())
eng = GpuEngine(d.tools, cfg, device=0)
cpu = CpuTranscoder()
mi = d.tools["bench_echoservice_echo"]

rng = random.Random(5)
if kind == "attrs":
    args = {"attrs": {f"p{j}": "x" * 500 for j in range(30)}}
elif kind == "scalar":
    args = {"f01String": "s" * 800, "f02Int32": 7}
else:
    args = wide_payload(rng, target_bytes=40000)

body = json.dumps({"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                   "params": {"name": "bench_echoservice_echo",
                              "arguments": args}}).encode()
print("encode...", flush=True)
enc, pbs = eng.encode_batch([body], mode=0)
assert enc[0]["status"] == 0, enc[0]
wire = pbs[0]
print(f"wire {len(wire)}B; decode mode0 (wg)...", flush=True)
dec, outs = eng.decode_batch([wire], [eng.tables.msg_index[mi.output_descriptor.full_name]], mode=0,
                             skip=[False])
print("dec status", dec[0]["status"], "out_len", dec[0]["out_len"], flush=True)
if os.environ["GGRMCP_WG_PHASES"] == "3" and dec[0]["status"] == 0:
    out = bytes(outs[0] or b"")
    resp = json.loads(out)
    inner = json.loads(resp["result"]["content"][0]["text"])
    oracle = json.loads(cpu.pb_to_json(mi.output_descriptor, wire))
    assert inner == oracle, (inner, oracle)
    print("oracle match", flush=True)
print("PROBE-OK", flush=True)

def diag_shapes():
    """kind=shapes: run the wg-decode test shapes through encode and print
    per-slot statuses (diagnosing validation rejections)."""
    sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "tests"))
    from test_gpu_wg_decode import _shapes

    shapes = _shapes()
    bodies = [json.dumps({"jsonrpc": "2.0", "id": i + 1, "method": "tools/call",
                          "params": {"name": "bench_echoservice_echo",
                                     "arguments": a}}).encode()
              for i, a in enumerate(shapes)]
    enc, pbs = eng.encode_batch(bodies, mode=0)
    for i, r in enumerate(enc):
        print(f"shape {i}: status={r['status']} aux={r['aux']} "
              f"err_pos={r['err_pos']} body_len={len(bodies[i])}", flush=True)
        if r["status"] != 0:
            ctx = bodies[i][max(0, r["err_pos"] - 40):r["err_pos"] + 40]
            print("   around err_pos:", ctx, flush=True)

if kind == "shapes":
    diag_shapes()
