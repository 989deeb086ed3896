"""Run the wide64 multi-slot decode repro through the CPU kernel build."""
import json
import random
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np

from examples.protos import ALL_FDPS
from ggrmcp_amd.descriptors.loader import build_pool, extract_method_infos
from ggrmcp_amd.engine.cpu_ref import CpuTranscoder
from ggrmcp_amd.engine.tables import compile_tables
from ggrmcp_amd.engine.batch import _offsets, DECODE_DTYPE, SLOT_DTYPE
from ggrmcp_amd.utils.synthetic import synthetic_fdp, wide_payload

sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "ggrmcp_amd" / "ops"))
import _hostsim

fdps = ALL_FDPS + [synthetic_fdp()]
pool = build_pool(fdps)
infos = {m.tool_name(): m for m in extract_method_infos(fdps, pool, compat_names=False)}
T = compile_tables(infos)
eng = _hostsim.HostEngine(T.msg_table, T.field_table, T.enum_table,
                          T.enum_values, T.tool_table, T.name_blob,
                          T.n_msgs, T.n_tools)
cpu = CpuTranscoder()
desc = pool.FindMessageTypeByName("bench.Wide64")
idx = T.msg_index["bench.Wide64"]

rng = random.Random(3)
p0 = wide_payload(rng)
n = 4
wires = [cpu.json_to_pb(desc, json.dumps(p0))] * n
lens = [len(w) for w in wires]
print("wire lens:", lens)
data = b"".join(wires)
resp_off = _offsets(lens)
scratch_off = _offsets([l * 8 + 1024 for l in lens], align=16)
final_off = _offsets([l * 16 + 2048 for l in lens], align=16)
dec, fin = eng.decode(data, resp_off, scratch_off, final_off,
                      np.full(n, idx, dtype=np.int32), None, 1)
recs = np.frombuffer(dec.tobytes(), dtype=DECODE_DTYPE)
for i in range(n):
    r = recs[i]
    print(f"slot {i}: status={int(r['status'])} off={int(r['out_off'])} len={int(r['out_len'])}")
    if r["status"] == 0:
        js = bytes(fin[r["out_off"]:r["out_off"]+r["out_len"]])
        ok = json.loads(js) == json.loads(cpu.pb_to_json(desc, wires[i]))
        print("   matches oracle:", ok)
