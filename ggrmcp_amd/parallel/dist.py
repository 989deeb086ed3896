"""DP sharding + collective state distribution over RCCL/xGMI.

The reference gateway is a single Go process whose shared state is an
``atomic.Pointer`` tool-map swap (aalobaidi/ggRMCP pkg/grpc/
discovery.go:21-30,122-127) and an in-process session cache
(pkg/session/manager.go).  The MI355X build runs ONE PROCESS PER GPU
(8 shards per node); this module is the native equivalent of that shared
state:

* ``shard_for_session`` — sessions hash stably onto a GPU shard, so each
  GPU owns its sessions end-to-end (no cross-GPU traffic on the data
  plane; the 7 xGMI p2p links carry control-plane state only).
* ``sync_discovery`` — rank 0 discovers the backend (reflection or
  descriptor set), serializes the descriptor snapshot, broadcasts it via
  ``torch.distributed`` (backend "nccl" IS RCCL on ROCm; a KB-scale
  broadcast over xGMI is latency-bound, so a single broadcast from rank 0
  beats any ring), and every rank rebuilds an identical tool map +
  compiled GPU tables — verified by an all-gathered FNV checksum.  The
  version barrier reproduces the reference's atomic-swap semantics
  across shards: no rank serves the new map until all ranks hold it.
* ``allreduce_stats`` — aggregate metrics (handler.go:367-376 exposes
  per-process stats; here /metrics reports whole-node sums).
"""

from __future__ import annotations

import logging
import os
from dataclasses import dataclass
from typing import Dict, Mapping, Optional

log = logging.getLogger("ggrmcp.parallel")

_FNV_OFFSET = 0xCBF29CE484222325
_FNV_PRIME = 0x100000001B3


def fnv1a64(data: bytes) -> int:
    h = _FNV_OFFSET
    for b in data:
        h ^= b
        h = (h * _FNV_PRIME) & 0xFFFFFFFFFFFFFFFF
    return h


def shard_for_session(session_id: str, world: int) -> int:
    """Stable session -> GPU shard mapping (Mcp-Session-Id affinity)."""
    if world <= 1:
        return 0
    return fnv1a64(session_id.encode()) % world


@dataclass
class ShardGroup:
    """One DP shard (process == GPU) of the gateway node."""

    rank: int
    world: int
    device: Optional[int] = None  # cuda device index; None = CPU/gloo
    _dist = None  # torch.distributed module when initialized

    @classmethod
    def from_env(cls, device: Optional[int] = None) -> "ShardGroup":
        """Initialize from torchrun env (RANK/WORLD_SIZE/MASTER_*).

        world==1 (or no env) -> degenerate group, no process group created.
        """
        world = int(os.environ.get("WORLD_SIZE", "1"))
        rank = int(os.environ.get("RANK", "0"))
        if world <= 1:
            return cls(rank=0, world=1, device=device)
        import torch
        import torch.distributed as dist

        use_gpu = device is not None and torch.cuda.is_available()
        if not dist.is_initialized():
            dist.init_process_group(backend="nccl" if use_gpu else "gloo")
        if use_gpu:
            torch.cuda.set_device(device)
        g = cls(rank=rank, world=world, device=device if use_gpu else None)
        g._dist = dist
        return g

    @classmethod
    def attach(cls, dist_module, device: Optional[int] = None) -> "ShardGroup":
        """Wrap an already-initialized torch.distributed process group."""
        g = cls(
            rank=dist_module.get_rank(),
            world=dist_module.get_world_size(),
            device=device,
        )
        g._dist = dist_module
        return g

    # -- collectives ---------------------------------------------------------

    def _tensor_device(self):
        return f"cuda:{self.device}" if self.device is not None else "cpu"

    def broadcast_blob(self, blob: Optional[bytes], src: int = 0) -> bytes:
        """Broadcast a byte blob from ``src`` to every shard.

        Two collectives: an 8-byte length, then the payload.  Small-payload
        broadcast from one root is the right xGMI shape (direct p2p fan-out
        inside RCCL; ring would be per-link bound for no benefit at KB
        scale).
        """
        if self.world == 1:
            assert blob is not None
            return blob
        import torch

        dev = self._tensor_device()
        n = torch.tensor(
            [len(blob) if blob is not None else 0], dtype=torch.int64, device=dev
        )
        self._dist.broadcast(n, src=src)
        size = int(n.item())
        if self.rank == src:
            t = torch.frombuffer(bytearray(blob), dtype=torch.uint8).to(dev)
        else:
            t = torch.empty(size, dtype=torch.uint8, device=dev)
        self._dist.broadcast(t, src=src)
        return bytes(t.cpu().numpy().tobytes())

    def verify_consistent(self, blob: bytes) -> bool:
        """All-gather an FNV checksum; True iff every shard holds identical
        bytes (the cross-shard analogue of the reference's single-process
        atomic map being trivially consistent)."""
        if self.world == 1:
            return True
        import torch

        dev = self._tensor_device()
        h = torch.tensor([fnv1a64(blob) - (1 << 63)], dtype=torch.int64, device=dev)
        out = [torch.zeros_like(h) for _ in range(self.world)]
        self._dist.all_gather(out, h)
        vals = {int(t.item()) for t in out}
        return len(vals) == 1

    def allreduce_stats(self, stats: Mapping[str, float]) -> Dict[str, float]:
        """Sum numeric stats across shards (whole-node /metrics)."""
        if self.world == 1:
            return dict(stats)
        import torch

        keys = sorted(stats.keys())
        dev = self._tensor_device()
        t = torch.tensor([float(stats[k]) for k in keys], dtype=torch.float64,
                         device=dev)
        self._dist.all_reduce(t, op=self._dist.ReduceOp.SUM)
        vals = t.cpu().tolist()
        return {k: vals[i] for i, k in enumerate(keys)}

    def barrier(self) -> None:
        if self.world > 1:
            self._dist.barrier()


def broadcast_blob(group: ShardGroup, blob: Optional[bytes], src: int = 0) -> bytes:
    return group.broadcast_blob(blob, src)


def allreduce_stats(group: ShardGroup, stats: Mapping[str, float]) -> Dict[str, float]:
    return group.allreduce_stats(stats)


def sync_discovery(discoverer, group: ShardGroup, src: int = 0) -> int:
    """Distribute rank ``src``'s discovered descriptor state to all shards.

    Rank src must have called ``discover()`` (or ``load_descriptor_blob``)
    already; the other ranks rebuild their tool maps from the broadcast
    snapshot, so every GPU compiles byte-identical transcode tables.
    Returns the published tools version.  The final barrier gives the
    all-shards-swap-together semantics of the reference's atomic.Pointer
    publish (discovery.go:122-127) — no shard serves the new map until
    every shard has it.
    """
    if group.world == 1:
        return discoverer.tools_version
    blob = discoverer.descriptor_blob() if group.rank == src else None
    blob = group.broadcast_blob(blob, src=src)
    if group.rank != src:
        discoverer.load_descriptor_blob(blob)
    if not group.verify_consistent(blob):  # pragma: no cover - defensive
        raise RuntimeError("descriptor snapshot diverged across shards")
    group.barrier()
    return discoverer.tools_version
