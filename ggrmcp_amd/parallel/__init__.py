"""Data-parallel sharding + RCCL state distribution."""

from .dist import (  # noqa: F401
    ShardGroup,
    allreduce_stats,
    broadcast_blob,
    shard_for_session,
    sync_discovery,
)
