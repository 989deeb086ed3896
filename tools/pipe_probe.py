"""Isolate the chunked-pipeline cost model (engine.cpp run_encode_chunked).

Times encode_batch on wide64-shaped payloads with ONE engine instance,
sweeping GGRMCP_PIPE_CHUNKS, so queue-contention effects from multiple
engine instances (profiles/streams_sweep.log) are excluded.  Run on a GPU
box:

    python tools/pipe_probe.py [--batch 64] [--payload 65536] [--iters 30]

Optionally rerun under GPU_MAX_HW_QUEUES=8 to test the HW-queue
multiplexing hypothesis for the chunk-linear slowdown.
"""

import argparse
import json
import os
import random
import statistics
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--payload", type=int, default=65536)
    ap.add_argument("--iters", type=int, default=30)
    ap.add_argument("--chunks", default="1,2,4,8")
    args = ap.parse_args()

    from google.protobuf import descriptor_pb2

    from examples.protos import ALL_FDPS
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer
    from ggrmcp_amd.config import Config
    from ggrmcp_amd.engine.batch import GpuPipeline
    from ggrmcp_amd.utils.synthetic import synthetic_fdp, wide_payload

    cfg = Config.default()
    cfg.gpu.pinned_pool_bytes = 1 << 30
    cfg.gpu.device_pool_bytes = 4 << 30
    cfg.gpu.streams = 1
    d = ServiceDiscoverer(cfg)
    fdset = descriptor_pb2.FileDescriptorSet()
    fdset.file.extend(ALL_FDPS + [synthetic_fdp()])
    d.load_descriptor_blob(fdset.SerializeToString())
    pipeline = GpuPipeline(d, cfg, device=0)

    rng = random.Random(11)
    bodies = []
    for i in range(args.batch):
        a = wide_payload(rng, target_bytes=args.payload)
        bodies.append(json.dumps(
            {"jsonrpc": "2.0", "id": i + 1, "method": "tools/call",
             "params": {"name": "bench_echoservice_echo",
                        "arguments": a}}).encode())
    total = sum(len(b) for b in bodies)
    print(f"batch={args.batch} payload~{args.payload} total={total}B "
          f"GPU_MAX_HW_QUEUES={os.environ.get('GPU_MAX_HW_QUEUES', '<unset>')}")

    for c in [int(x) for x in args.chunks.split(",")]:
        os.environ["GGRMCP_PIPE_CHUNKS"] = str(c)
        os.environ["GGRMCP_PIPE_MIN"] = "1"
        # warmup
        for _ in range(3):
            pipeline.engine.encode_batch(bodies, mode=0)
        ts = []
        for _ in range(args.iters):
            t0 = time.perf_counter()
            pipeline.engine.encode_batch(bodies, mode=0)
            ts.append((time.perf_counter() - t0) * 1e3)
        ts.sort()
        print(f"chunks={c}: encode p50={statistics.median(ts):.3f}ms "
              f"min={ts[0]:.3f} p90={ts[int(len(ts)*0.9)]:.3f} "
              f"({total / statistics.median(ts) / 1e6:.2f} GB/s)")
    for k in ("GGRMCP_PIPE_CHUNKS", "GGRMCP_PIPE_MIN"):
        os.environ.pop(k, None)
    d.close()


if __name__ == "__main__":
    main()
