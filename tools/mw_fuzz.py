"""One-shot large differential fuzz of the multi-wave structural scan.

Generates several hundred payloads (wide_payload shapes, adversarial
mutations: structural chars in strings, escape runs, boundary-straddling
tokens, random truncations/corruptions) and requires the mw-forced
encode to match the serial wg scanner byte-for-byte on every slot.

    python tools/mw_fuzz.py [--rounds 8] [--batch 48]
"""

import argparse
import json
import os
import random
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def bodies_for(rng, batch):
    from ggrmcp_amd.utils.synthetic import wide_payload

    out = []
    for i in range(batch):
        kind = rng.randrange(6)
        if kind == 0:
            args = wide_payload(rng, target_bytes=rng.choice(
                [9000, 17000, 33000, 64000]))
        elif kind == 1:  # structural chars inside strings
            evil = '{"a":[1,2],\\}' + "{[,]}" * rng.randint(5, 60)
            args = {"attrs": {f"k{j}": (evil * 3)[: rng.randint(100, 1000)]
                              for j in range(rng.randint(10, 30))}}
        elif kind == 2:  # escape runs at varying alignments
            v = "x" * rng.randint(200, 300) + "\\" * rng.randint(1, 4)
            args = {"attrs": {f"b{j}": v + "y" * (j % 11)
                              for j in range(rng.randint(12, 40))}}
        elif kind == 3:  # many small members
            args = {"attrs": {f"m{j:04d}": str(rng.random())
                              for j in range(rng.randint(150, 300))}}
        elif kind == 4:  # nested + unicode
            nest = {"name": "leaf" + "é中" * rng.randint(1, 30)}
            for _ in range(rng.randint(1, 9)):
                nest = {"child": nest, "name": "n" * rng.randint(1, 60)}
            args = {"f40Node": nest,
                    "attrs": {f"n{j}": "m" * 700
                              for j in range(rng.randint(8, 20))}}
        else:  # big repeated-message runs (chunk_arr candidates)
            args = {"items": [
                {"key": "k" * rng.randint(1, 400),
                 "value": str(rng.randint(0, 10**9)),
                 "weight": rng.random()}
                for _ in range(rng.randint(20, 80))]}
        body = json.dumps(
            {"jsonrpc": "2.0", "id": i + 1, "method": "tools/call",
             "params": {"name": "bench_echoservice_echo",
                        "arguments": args}}).encode()
        # occasional corruption: truncate or flip a byte (both paths must
        # agree on the error result too)
        r = rng.random()
        if r < 0.06 and len(body) > 40:
            body = body[: rng.randrange(30, len(body))]
        elif r < 0.12:
            k = rng.randrange(20, len(body))
            body = body[:k] + bytes([rng.randrange(32, 127)]) + body[k + 1:]
        out.append(body)
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rounds", type=int, default=8)
    ap.add_argument("--batch", type=int, default=48)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()

    from google.protobuf import descriptor_pb2

    from examples.protos import ALL_FDPS
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer
    from ggrmcp_amd.config import Config
    from ggrmcp_amd.engine.batch import GpuPipeline
    from ggrmcp_amd.utils.synthetic import synthetic_fdp

    cfg = Config.default()
    cfg.gpu.pinned_pool_bytes = 1 << 30
    cfg.gpu.device_pool_bytes = 4 << 30
    cfg.gpu.streams = 1
    d = ServiceDiscoverer(cfg)
    fdset = descriptor_pb2.FileDescriptorSet()
    fdset.file.extend(ALL_FDPS + [synthetic_fdp()])
    d.load_descriptor_blob(fdset.SerializeToString())
    pipeline = GpuPipeline(d, cfg, device=0)

    total = 0
    for rd in range(args.rounds):
        rng = random.Random(args.seed * 10007 + rd)
        bodies = bodies_for(rng, args.batch)
        os.environ["GGRMCP_WG_ENC_MIN"] = "2048"
        os.environ["GGRMCP_MW_SCAN"] = "0"
        ser = pipeline.engine.encode_batch(bodies, mode=0)
        os.environ["GGRMCP_MW_SCAN"] = "1"
        mw = pipeline.engine.encode_batch(bodies, mode=0)
        for k in ("GGRMCP_WG_ENC_MIN", "GGRMCP_MW_SCAN"):
            os.environ.pop(k, None)
        for i in range(len(bodies)):
            for f in ("status", "tool_idx", "err_pos", "aux", "id_len"):
                assert ser[0][i][f] == mw[0][i][f], (
                    f"round {rd} slot {i} field {f}: "
                    f"{ser[0][i][f]} != {mw[0][i][f]}")
            assert ser[1][i] == mw[1][i], f"round {rd} slot {i} wire diverged"
        total += len(bodies)
        print(f"round {rd}: {len(bodies)} ok")
    print(f"mw fuzz clean: {total} payloads byte-identical")
    d.close()


if __name__ == "__main__":
    main()
