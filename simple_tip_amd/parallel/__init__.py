"""Data-parallel sharding over RCCL/xGMI (torch.distributed)."""

from .dist import (
    allgather_rows,
    allreduce_max_scalar,
    barrier,
    get_rank,
    get_world_size,
    init_from_env,
    is_initialized,
    shard_slice,
)

__all__ = [
    "init_from_env",
    "is_initialized",
    "get_rank",
    "get_world_size",
    "shard_slice",
    "allgather_rows",
    "allreduce_max_scalar",
    "barrier",
]
