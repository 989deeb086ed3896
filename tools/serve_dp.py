"""Multi-rank DP serving launcher: one gateway process per GPU, ONE port.

The deployment shape for the 8-GPU node (SURVEY §5 "sessions are sharded
DP so each GPU owns its sessions end-to-end"): every rank runs the full
native gateway (C++ HTTP reactors -> GPU pipeline -> native gRPC
transport) bound to the SAME TCP port via SO_REUSEPORT; the kernel
balances client connections across ranks, and rank 0 broadcasts the
descriptor snapshot over RCCL/gloo so all ranks serve an identical tool
set (parallel/dist.sync_discovery).  No reference equivalent — the
reference is a single Go process (cmd/grmcp/main.go).

Launch (8 GPUs):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 tools/serve_dp.py \
        --port 50053 --descriptor svc.binpb --grpc-host ... [--no-gpu]

Each rank reads RANK/LOCAL_RANK/WORLD_SIZE from the environment.  With
--no-gpu (or no CUDA) it serves the CPU reference pipeline — which is how
the world-2 CPU test exercises this file end to end.

Session affinity: SO_REUSEPORT balances CONNECTIONS, but session state
(counters, blocks, rate windows) lives in ONE /dev/shm-backed C++ table
mapped by every rank (ops/csrc/session_table.h, wired below via
config.session.shared_table_path), so a client that reconnects and lands
on a different rank keeps its session context — the round-1 affinity
caveat is closed.  Pass --no-shared-sessions to
revert to per-rank state.
"""

from __future__ import annotations

import argparse
import os
import signal
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from ggrmcp_amd.backend.discovery import ServiceDiscoverer  # noqa: E402
from ggrmcp_amd.config import Config  # noqa: E402


def parse_args():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=50053)
    ap.add_argument("--grpc-host", default="localhost")
    ap.add_argument("--grpc-port", type=int, default=50051)
    ap.add_argument("--uds", default="", help="backend unix socket")
    ap.add_argument("--descriptor", default="", help=".binpb FileDescriptorSet")
    ap.add_argument("--no-gpu", action="store_true")
    ap.add_argument("--no-shared-sessions", action="store_true",
                    help="per-rank session state (round-1 behavior)")
    ap.add_argument("--session-table", default="",
                    help="explicit shared session table path "
                         "(default: /dev/shm/ggrmcp_sessions_<port>)")
    ap.add_argument("--run-seconds", type=float, default=0.0,
                    help="exit after N seconds (0 = run until SIGTERM)")
    return ap.parse_args()


def main() -> None:
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    import torch

    use_gpu = not args.no_gpu and torch.cuda.is_available()
    dev = (local_rank % max(1, torch.cuda.device_count())) if use_gpu else 0

    cfg = Config.default()
    cfg.grpc.host, cfg.grpc.port = args.grpc_host, args.grpc_port
    cfg.grpc.uds = args.uds
    cfg.gpu.enabled = use_gpu
    cfg.server.http_port = args.port
    cfg.server.reuse_port = world > 1  # all ranks share the port
    if world > 1 and not args.no_shared_sessions:
        # one session table for the whole node: reconnects keep their
        # session no matter which rank SO_REUSEPORT lands them on
        cfg.session.shared_table_path = args.session_table or (
            f"/dev/shm/ggrmcp_sessions_{args.port}"
        )

    dist = None
    shard_group = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        backend = os.environ.get("GGRMCP_DIST_BACKEND") or (
            "nccl" if use_gpu else "gloo"
        )
        if use_gpu:
            torch.cuda.set_device(dev)
        dist.init_process_group(backend=backend)
        from ggrmcp_amd.parallel.dist import ShardGroup

        shard_group = ShardGroup.attach(
            dist, device=dev if (use_gpu and backend == "nccl") else None
        )
        if cfg.session.shared_table_path:
            # rank 0 clears any stale table from a previous run BEFORE any
            # rank maps it (barrier orders the unlink against the opens)
            if rank == 0 and os.path.exists(cfg.session.shared_table_path):
                os.unlink(cfg.session.shared_table_path)
            dist.barrier()

    discoverer = ServiceDiscoverer(cfg)
    # rank 0 discovers (descriptor file, or live reflection) and broadcasts
    # the framed snapshot so every rank serves IDENTICAL tools
    # (discovery.go:122-127's atomic swap, node-wide)
    if rank == 0 or shard_group is None:
        if args.descriptor:
            discoverer.load_descriptor_blob(Path(args.descriptor).read_bytes())
        else:
            discoverer.connect(timeout_s=30)
            discoverer.discover()
    if shard_group is not None:
        from ggrmcp_amd.parallel.dist import sync_discovery

        sync_discovery(discoverer, shard_group, src=0)
    discoverer.connections[0].connect(timeout_s=15)

    from ggrmcp_amd.server.native_http import (
        CpuBatchPipeline, NativeHTTPGateway,
    )

    if use_gpu:
        from ggrmcp_amd.engine.batch import GpuPipeline, build_wire_clients

        pipeline = GpuPipeline(discoverer, cfg, device=dev,
                               wire_clients=build_wire_clients(discoverer, cfg))
    else:
        pipeline = CpuBatchPipeline(discoverer)

    gw = NativeHTTPGateway(pipeline, discoverer, cfg, host=args.host,
                           port=args.port)
    port = gw.start()
    print(f"[serve_dp] rank {rank}/{world} serving on {args.host}:{port} "
          f"({'gpu:%d' % dev if use_gpu else 'cpu'})", flush=True)

    stop = {"flag": False}
    signal.signal(signal.SIGTERM, lambda *_: stop.update(flag=True))
    signal.signal(signal.SIGINT, lambda *_: stop.update(flag=True))
    deadline = time.time() + args.run_seconds if args.run_seconds > 0 else None
    try:
        while not stop["flag"]:
            if deadline is not None and time.time() >= deadline:
                break
            time.sleep(0.2)
    finally:
        gw.stop(drain_s=cfg.server.shutdown_drain_s)
        if dist is not None:
            dist.destroy_process_group()


if __name__ == "__main__":
    main()
