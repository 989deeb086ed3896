"""Per-shape kernel timing: which payload shapes cost what on the GPU."""
import json, sys, time
from pathlib import Path
ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))

from ggrmcp_amd.utils.protobuild import FileBuilder
from ggrmcp_amd.descriptors.loader import build_pool, extract_method_infos
from ggrmcp_amd.engine.batch import GpuEngine
from ggrmcp_amd.engine.cpu_ref import CpuTranscoder
from ggrmcp_amd.utils.synthetic import synthetic_fdp

fb = FileBuilder("p/probe.proto", "p")
fb.message("Str").field("s", 1, "string").done()
fb.message("Map").map_field("m", 1, "string", "string").done()
fb.message("Pack").field("d", 1, "double", repeated=True).done()
fb.message("Ints").field("v", 1, "int64", repeated=True).done()
(fb.service("S").method("M", "Str", "Str").method("M2", "Map", "Map")
   .method("M3", "Pack", "Pack").method("M4", "Ints", "Ints").done())
fdp = fb.build()
fdps = [fdp, synthetic_fdp()]
pool = build_pool(fdps)
infos = {m.tool_name(): m for m in extract_method_infos(fdps, pool, compat_names=False)}
eng = GpuEngine(infos, device=0)
cpu = CpuTranscoder()

# the real wide64 payload + components thereof
import json as _json
from ggrmcp_amd.utils.synthetic import synthetic_fdp, wide_payload
import random as _random

_rngW = _random.Random(7)
_wp = wide_payload(_rngW, target_bytes=64 * 1024)
_scalars = {k: v for k, v in _wp.items() if k.startswith("f")}
_attrs_only = {"attrs": _wp["attrs"]}

CASES = {
    "str64k": ("p.Str", {"s": "x" * 65536}),
    "str1k": ("p.Str", {"s": "x" * 1024}),
    "map64x1k": ("p.Map", {"m": {f"k{i:02d}": "v" * 1000 for i in range(64)}}),
    "map1024x64": ("p.Map", {"m": {f"key{i:04d}": "v" * 56 for i in range(1024)}}),
    "pack8k_doubles": ("p.Pack", {"d": [i * 1.5 for i in range(8192)]}),
    "ints4k": ("p.Ints", {"v": [str(i * 7) for i in range(4096)]}),
    "wide_actual": ("bench.Wide64", _wp),
    "wide_scalars": ("bench.Wide64", _scalars),
    "wide_attrs": ("bench.Wide64", _attrs_only),
}

BATCH = 128
for name, (msg, payload) in CASES.items():
    desc = pool.FindMessageTypeByName(msg)
    text = json.dumps(payload)
    idx = eng.tables.msg_index[msg]
    wire = cpu.json_to_pb(desc, text)
    # decode timing
    eng.decode_batch([wire] * BATCH, [idx] * BATCH, mode=1)  # warmup
    eng.stats.decode_ns = 0
    for _ in range(3):
        dec, outs = eng.decode_batch([wire] * BATCH, [idx] * BATCH, mode=1)
    ok = int(dec[0]["status"])
    dec_us = eng.stats.decode_ns / 3 / 1e3
    # encode timing
    eng.encode_batch([text.encode()] * BATCH, mode=1,
                     msg_indices=[idx] * BATCH, enforce=False)  # warmup
    eng.stats.encode_ns = 0
    for _ in range(3):
        enc, pbs = eng.encode_batch([text.encode()] * BATCH, mode=1,
                                    msg_indices=[idx] * BATCH, enforce=False)
    eok = int(enc[0]["status"])
    enc_us = eng.stats.encode_ns / 3 / 1e3
    per_kb_dec = dec_us / (len(wire) / 1024)
    per_kb_enc = enc_us / (len(text) / 1024)
    print(f"{name:16s} wire={len(wire):7d}B json={len(text):7d}B "
          f"dec={dec_us:9.0f}us/b{BATCH} ({per_kb_dec:6.1f}us/KB-req) st={ok} "
          f"enc={enc_us:9.0f}us ({per_kb_enc:6.1f}us/KB-req) st={eok}", flush=True)
