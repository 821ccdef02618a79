"""torch.distributed helpers for data-parallel TIP sharding.

The reference has NO inter-worker communication (SURVEY.md §2.4 — process
pools sharing only the filesystem). Here test inputs shard across the 8
GPUs of one MI355X node; per-input priority scores are all-gathered over
RCCL/xGMI (backend "nccl" IS RCCL on ROCm). Messages are kilobytes — these
collectives are latency-bound, the compute (pairwise kernels) dominates.
Works identically with the gloo backend on CPU (world_size>1 unit tests).
"""

import datetime
import os
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist


def init_from_env(backend: Optional[str] = None) -> Tuple[int, int, torch.device]:
    """Initialise from torchrun env vars; returns (rank, world, device).

    Single-process (no WORLD_SIZE or WORLD_SIZE=1) leaves torch.distributed
    uninitialised and returns rank 0.
    """
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        dev = torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")
        return 0, 1, dev
    if backend is None:
        # TIP_DIST_BACKEND=gloo lets a multi-rank run share one GPU (RCCL
        # refuses duplicate devices) — used for in-lease validation
        backend = os.environ.get("TIP_DIST_BACKEND") or (
            "nccl" if torch.cuda.is_available() else "gloo"
        )
    local_rank = int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))
    if torch.cuda.is_available():
        dev_id = local_rank % torch.cuda.device_count()
        torch.cuda.set_device(dev_id)
        dev = torch.device(f"cuda:{dev_id}")
    else:
        dev = torch.device("cpu")
    if not dist.is_initialized():
        dist.init_process_group(
            backend=backend, timeout=datetime.timedelta(seconds=300)
        )
    return dist.get_rank(), dist.get_world_size(), dev


def is_initialized() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_initialized() else 1


def barrier():
    if is_initialized():
        dist.barrier()


def shard_slice(n: int, rank: Optional[int] = None, world: Optional[int] = None) -> slice:
    """Contiguous shard of [0, n) for this rank (first shards get the
    remainder, matching torch.tensor_split)."""
    rank = get_rank() if rank is None else rank
    world = get_world_size() if world is None else world
    base, rem = divmod(n, world)
    start = rank * base + min(rank, rem)
    return slice(start, start + base + (1 if rank < rem else 0))


def _staging_device(t: torch.Tensor):
    """Where a tensor must live for this backend's collectives: nccl (RCCL)
    needs device tensors; gloo implements all_gather only on host. Returns
    (tensor staged for the backend, original device or None if unmoved)."""
    backend = dist.get_backend()
    if backend == "nccl" and not t.is_cuda:
        return t.cuda(), t.device
    if backend == "gloo" and t.is_cuda:
        return t.cpu(), t.device
    return t, None


def gather_tensors(t: torch.Tensor) -> List[torch.Tensor]:
    """all_gather equal-shaped tensors with backend/device staging; results
    come back on the input's device, in rank order."""
    world = get_world_size()
    staged, home = _staging_device(t)
    out = [torch.empty_like(staged) for _ in range(world)]
    dist.all_gather(out, staged.contiguous())
    if home is not None:
        out = [o.to(home) for o in out]
    return out


def allgather_rows(local: torch.Tensor, n_total: int) -> torch.Tensor:
    """All-gather row shards produced by :func:`shard_slice` into the full
    [n_total, ...] tensor (identical on every rank)."""
    if not is_initialized():
        return local
    world = get_world_size()
    pad = (n_total + world - 1) // world  # equal-size buffers (RCCL-safe)
    buf = torch.zeros((pad,) + tuple(local.shape[1:]), dtype=local.dtype,
                      device=local.device)
    buf[: local.shape[0]] = local
    shards = gather_tensors(buf)
    parts = []
    for r in range(world):
        s = shard_slice(n_total, r, world)
        parts.append(shards[r][: s.stop - s.start])
    return torch.cat(parts, dim=0)


def allreduce_max_scalar(value: float, device) -> float:
    """MAX all-reduce of a python float (slowest-rank timing)."""
    if not is_initialized():
        return value
    t = torch.tensor([value], dtype=torch.float64, device=device)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())
