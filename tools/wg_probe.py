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

def diag_shape_diff(which: int):
    """kind=shapediffN: decode test shape N and print first oracle diff."""
    sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "tests"))
    from test_gpu_wg_decode import _shapes

    args = _shapes()[which]
    wire = cpu.json_to_pb(mi.input_descriptor, json.dumps(args))
    out_idx = eng.tables.msg_index[mi.output_descriptor.full_name]
    dec, outs = eng.decode_batch([wire], [out_idx], mode=1)
    print("dec status", dec[0]["status"], flush=True)
    if dec[0]["status"] != 0:
        return
    kern = json.loads(outs[0])
    orac = json.loads(cpu.pb_to_json(mi.output_descriptor, wire))
    for k in orac:
        if kern.get(k) != orac[k]:
            a, b = kern.get(k), orac[k]
            if isinstance(a, list):
                for j, (x, y) in enumerate(zip(a, b)):
                    if x != y:
                        print(f"field {k}[{j}]:\n  kern {x}\n  orac {y}", flush=True)
                        return
            print(f"field {k}:\n  kern {str(a)[:200]}\n  orac {str(b)[:200]}", flush=True)
            return
    print("no diff", flush=True)

if kind.startswith("shapediff"):
    diag_shape_diff(int(kind[len("shapediff"):]))

def diag_doubles():
    """kind=dbl: minimal double-emission check on this device."""
    desc = mi.input_descriptor.fields_by_name["nested"].message_type
    idx = eng.tables.msg_index["bench.Inner"]
    vals = [301/3, 1/3, 3333333333333333.5, 0.1+0.2, 7.479800121866815e-11,
            2.4392533358425466e-50, 223355.53145190026, 6.098280709992834e+16]
    wires = [cpu.json_to_pb(desc, json.dumps({"weight": v})) for v in vals]
    dec, outs = eng.decode_batch(wires, [idx] * len(wires), mode=1)
    for v, r, o in zip(vals, dec, outs):
        ok = r["status"] == 0 and json.loads(o).get("weight") == v
        print(("OK " if ok else "BAD"), repr(v), "->",
              o if r["status"] == 0 else f"status={r['status']}", flush=True)

if kind == "dbl":
    diag_doubles()

def diag_wg_diff(which: int):
    """kind=wgdiffN: shape N through encode(mode0)+decode(mode0) — the
    workgroup kernel path — diffed against protojson."""
    sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "tests"))
    from test_gpu_wg_decode import _shapes

    args = _shapes()[which]
    body = json.dumps({"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                       "params": {"name": "bench_echoservice_echo",
                                  "arguments": args}}).encode()
    enc, pbs = eng.encode_batch([body], mode=0)
    assert enc[0]["status"] == 0, enc[0]
    wire = pbs[0]
    out_idx = eng.tables.msg_index[mi.output_descriptor.full_name]
    dec, outs = eng.decode_batch([wire], [out_idx], mode=0, skip=[False])
    print("wire", len(wire), "dec status", dec[0]["status"], flush=True)
    if dec[0]["status"] != 0:
        return
    resp = json.loads(outs[0])
    kern = json.loads(resp["result"]["content"][0]["text"])
    orac = json.loads(cpu.pb_to_json(mi.output_descriptor, wire))
    for k in sorted(set(orac) | set(kern)):
        if kern.get(k) != orac.get(k):
            a, b = kern.get(k), orac.get(k)
            if isinstance(a, list) and isinstance(b, list):
                print(f"len {len(a)} vs {len(b)}", flush=True)
                for j, (x, y) in enumerate(zip(a, b)):
                    if x != y:
                        print(f"field {k}[{j}]:\n  kern {x}\n  orac {y}", flush=True)
                        return
            print(f"field {k}:\n  kern {str(a)[:300]}\n  orac {str(b)[:300]}", flush=True)
            return
    print("no diff (wg path)", flush=True)

if kind.startswith("wgdiff"):
    diag_wg_diff(int(kind[len("wgdiff"):]))

def diag_batch6():
    """kind=batch6: all test shapes in ONE mode-0 batch (multi-block wg)."""
    sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "tests"))
    from test_gpu_wg_decode import _shapes

    shapes = _shapes()
    bodies = [json.dumps({"jsonrpc": "2.0", "id": i + 1, "method": "tools/call",
                          "params": {"name": "bench_echoservice_echo",
                                     "arguments": a}}).encode()
              for i, a in enumerate(shapes)]
    enc, pbs = eng.encode_batch(bodies, mode=0)
    out_idx = eng.tables.msg_index[mi.output_descriptor.full_name]
    dec, outs = eng.decode_batch(pbs, [out_idx] * len(pbs), mode=0,
                                 skip=[False] * len(pbs))
    for i, (w, r, o) in enumerate(zip(pbs, dec, outs)):
        if r["status"] != 0:
            print(f"slot {i}: dec status {r['status']} (host fallback)", flush=True)
            continue
        resp = json.loads(o)
        kern = json.loads(resp["result"]["content"][0]["text"])
        orac = json.loads(cpu.pb_to_json(mi.output_descriptor, w))
        if kern == orac:
            print(f"slot {i}: ok", flush=True)
            continue
        for k in sorted(set(orac) | set(kern)):
            if kern.get(k) != orac.get(k):
                a, b = kern.get(k), orac.get(k)
                if isinstance(a, list) and isinstance(b, list):
                    for j, (x, y) in enumerate(zip(a, b)):
                        if x != y:
                            print(f"slot {i} field {k}[{j}]:\n  kern {x}\n  orac {y}", flush=True)
                            break
                    else:
                        print(f"slot {i} field {k}: len {len(a)} vs {len(b)}", flush=True)
                else:
                    print(f"slot {i} field {k}:\n  kern {str(a)[:200]}\n  orac {str(b)[:200]}", flush=True)
                break

if kind == "batch6":
    diag_batch6()

def enc_time():
    """kind=enctime: time encode_batch alone on big wide payloads."""
    import time
    from ggrmcp_amd.utils.synthetic import wide_payload, jsonrpc_body

    rng = random.Random(5)
    bodies = [jsonrpc_body("bench_echoservice_echo",
                           wide_payload(rng, target_bytes=64 * 1024), i)
              for i in range(64)]
    eng.encode_batch(bodies, mode=0)  # warm
    t0 = time.perf_counter()
    for _ in range(20):
        eng.encode_batch(bodies, mode=0)
    dt = (time.perf_counter() - t0) / 20
    print(f"encode_batch(64x64KB): {dt*1e3:.2f} ms/iter", flush=True)

if kind == "enctime":
    enc_time()
