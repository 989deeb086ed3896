"""Synthetic proto services + payload generators for benchmarks and tests.

BASELINE.json configs name: hello-service SayHello with 1 KB JSON payloads
(config 2) and a synthetic 64-field nested proto with 64 KB payloads
(config 3).  Payload generation is seeded and deterministic.
"""

from __future__ import annotations

import json
import random
import string
from typing import Dict, List

from .protobuild import FileBuilder

_SCALAR_CYCLE = [
    "string", "int32", "int64", "double", "bool", "uint32", "uint64", "float",
    "sint32", "sint64", "fixed32", "fixed64", "sfixed32", "sfixed64",
]


def synthetic_fdp(n_fields: int = 64, package: str = "bench"):
    """bench.proto: Wide64 (n_fields mixed scalars + nested + repeated +
    enum + map) and an EchoService echoing it.  ``package`` distinguishes
    per-backend copies in centralized-gateway mode (BASELINE config 5)."""
    fb = FileBuilder(f"{package}/bench.proto", package)
    fb.enum("Level", [("LEVEL_UNSET", 0), ("LOW", 1), ("MID", 2), ("HIGH", 3)])
    inner = fb.message("Inner")
    inner.field("key", 1, "string")
    inner.field("value", 2, "int64")
    inner.field("weight", 3, "double")
    inner.done()
    msg = fb.message("Wide64")
    num = 1
    for i in range(n_fields - 4):
        kind = _SCALAR_CYCLE[i % len(_SCALAR_CYCLE)]
        msg.field(f"f{num:02d}_{kind}", num, kind)
        num += 1
    msg.field("nested", num, "message", message="Inner")
    num += 1
    msg.field("items", num, "message", message="Inner", repeated=True)
    num += 1
    msg.field("level", num, "enum", enum="Level")
    num += 1
    msg.map_field("attrs", num, "string", "string")
    msg.done()
    (
        fb.service("EchoService")
        .method("Echo", "Wide64", "Wide64")
        .method("StreamEcho", "Wide64", "Wide64", server_streaming=True)
        .done()
    )
    return fb.build()


def wide_payload(rng: random.Random, n_fields: int = 64, target_bytes: int = 0) -> Dict:
    """JSON arguments for Wide64.  With target_bytes, pads strings so the
    JSON text lands near that size."""
    out: Dict = {}
    num = 1
    for i in range(n_fields - 4):
        kind = _SCALAR_CYCLE[i % len(_SCALAR_CYCLE)]
        name = f"f{num:02d}{''.join(p.capitalize() for p in [kind])}"
        # json_name of f01_string is f01String
        if kind == "string":
            out[name] = "".join(rng.choices(string.ascii_letters, k=24))
        elif kind in ("double", "float"):
            out[name] = round(rng.uniform(-1000, 1000), 3)
        elif kind == "bool":
            out[name] = rng.random() < 0.5
        elif kind in ("int64", "sint64", "sfixed64"):
            out[name] = str(rng.randint(-(2**62), 2**62))
        elif kind in ("uint64", "fixed64"):
            out[name] = str(rng.randint(0, 2**63))
        elif kind in ("uint32", "fixed32"):
            out[name] = rng.randint(0, 2**31)
        else:
            out[name] = rng.randint(-(2**31), 2**31 - 1)
        num += 1
    out["nested"] = {"key": "n", "value": "42", "weight": 1.5}
    out["items"] = [
        {"key": f"i{j}", "value": str(j * 7), "weight": j / 2} for j in range(4)
    ]
    out["level"] = rng.choice(["LOW", "MID", "HIGH"])
    out["attrs"] = {f"k{j}": "v" * 8 for j in range(4)}
    if target_bytes:
        cur = len(json.dumps(out))
        if cur < target_bytes:
            pad = target_bytes - cur - 20
            chunks = max(1, pad // 1000)
            out["attrs"].update(
                {
                    f"pad{j}": "".join(rng.choices(string.ascii_letters, k=min(1000, pad // chunks)))
                    for j in range(chunks)
                }
            )
    return out


def hello_payload(rng: random.Random, target_bytes: int = 1024) -> Dict:
    """SayHello arguments padded to ~target_bytes of JSON (BASELINE config 2:
    1 KB payloads)."""
    base = 20
    return {"name": "".join(rng.choices(string.ascii_letters, k=max(1, target_bytes - base)))}


def jsonrpc_body(tool: str, args: Dict, rid) -> bytes:
    return json.dumps(
        {
            "jsonrpc": "2.0",
            "id": rid,
            "method": "tools/call",
            "params": {"name": tool, "arguments": args},
        },
        ensure_ascii=False,
    ).encode()
