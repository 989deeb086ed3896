"""Root-cause probe for the multi-instance HIP submission contention
(profiles/streams_sweep.log: same per-batch encode work inflates 9.3 ms ->
69 ms device-span going 1 -> 4 engine instances).

N threads, each looping encode_batch on its OWN engine instance,
serving-shaped bodies (1 KB x 256).  Reports per-batch p50 span vs N.
If the inflation reproduces, rerun under candidate env knobs:

    python tools/contention_probe.py
    GPU_MAX_HW_QUEUES=16 python tools/contention_probe.py
    AMD_DIRECT_DISPATCH=0 python tools/contention_probe.py
    HSA_MAX_QUEUES=16 python tools/contention_probe.py

Each engine owns 3 streams (compute + chunk-H2D + chunk-D2H), so N=6
engines = 18 streams; ROCm multiplexes streams onto GPU_MAX_HW_QUEUES
(default 4) HW queues per process.
"""

import argparse
import json
import os
import statistics
import sys
import threading
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--payload", type=int, default=1024)
    ap.add_argument("--iters", type=int, default=60)
    ap.add_argument("--engines", default="1,2,4,6")
    args = ap.parse_args()

    from google.protobuf import descriptor_pb2

    from examples.protos import ALL_FDPS
    from ggrmcp_amd.backend.discovery import ServiceDiscoverer
    from ggrmcp_amd.config import Config
    from ggrmcp_amd.engine.batch import GpuPipeline
    from ggrmcp_amd.utils.synthetic import synthetic_fdp

    cfg = Config.default()
    cfg.gpu.streams = 1
    d = ServiceDiscoverer(cfg)
    fdset = descriptor_pb2.FileDescriptorSet()
    fdset.file.extend(ALL_FDPS + [synthetic_fdp()])
    d.load_descriptor_blob(fdset.SerializeToString())

    body = json.dumps(
        {"jsonrpc": "2.0", "id": 1, "method": "tools/call",
         "params": {"name": "bench_echoservice_echo",
                    "arguments": {"f01String": "x" * max(8, args.payload - 120),
                                  "f02Int32": 7}}}).encode()
    bodies = [body] * args.batch

    max_e = max(int(x) for x in args.engines.split(","))
    pipes = [GpuPipeline(d, cfg, device=0) for _ in range(max_e)]
    for p in pipes:  # warmup each engine's arenas
        p.engine.encode_batch(bodies, mode=0)

    print(f"batch={args.batch}x{args.payload}B iters={args.iters} "
          f"GPU_MAX_HW_QUEUES={os.environ.get('GPU_MAX_HW_QUEUES', '<unset>')} "
          f"AMD_DIRECT_DISPATCH={os.environ.get('AMD_DIRECT_DISPATCH', '<unset>')}")

    for ne in [int(x) for x in args.engines.split(",")]:
        spans = [[] for _ in range(ne)]

        def worker(k):
            eng = pipes[k].engine
            for _ in range(args.iters):
                t0 = time.perf_counter()
                eng.encode_batch(bodies, mode=0)
                spans[k].append((time.perf_counter() - t0) * 1e3)

        ths = [threading.Thread(target=worker, args=(k,)) for k in range(ne)]
        t0 = time.perf_counter()
        for t in ths:
            t.start()
        for t in ths:
            t.join()
        wall = time.perf_counter() - t0
        allspans = sorted(s for ss in spans for s in ss)
        p50 = statistics.median(allspans)
        p90 = allspans[int(len(allspans) * 0.9)]
        total = ne * args.iters
        print(f"engines={ne}: per-batch p50={p50:.3f}ms p90={p90:.3f}ms "
              f"wall={wall * 1e3:.0f}ms batches/s={total / wall:.0f}")
    d.close()


if __name__ == "__main__":
    main()
