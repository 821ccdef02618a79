"""Data-parallel sharding over RCCL/xGMI (torch.distributed)."""

from .dist import (
    allgather_rows,
    allreduce_max_scalar,
    barrier,
    gather_tensors,
    get_rank,
    get_world_size,
    init_from_env,
    is_initialized,
    shard_slice,
)
from .sharded import (
    allreduce_bitmap_or,
    allreduce_minmax,
    allreduce_welford,
    shard_rows,
    sharded_kde_logsumexp,
    sharded_rowmin_l2,
)

__all__ = [
    "init_from_env",
    "is_initialized",
    "get_rank",
    "get_world_size",
    "shard_slice",
    "shard_rows",
    "allgather_rows",
    "gather_tensors",
    "allreduce_max_scalar",
    "allreduce_minmax",
    "allreduce_welford",
    "allreduce_bitmap_or",
    "sharded_rowmin_l2",
    "sharded_kde_logsumexp",
    "barrier",
]
