"""Train-AT-sharded reductions over RCCL/xGMI.

The workload's large axis is the training-AT dimension of DSA/KDE
(SURVEY.md §5 "long-context" analogue: up to 60k x 2304 floats). These
helpers shard that axis across ranks: every rank holds a contiguous
train-AT shard, scores the (replicated) test inputs against its shard with
the local MFMA kernel, and the partials combine with small collectives —
all_gather + deterministic rank-ordered merges, so 1-GPU and N-GPU results
are bitwise identical.
"""

from typing import Tuple

import torch
import torch.distributed as dist

from .. import ops
from .dist import (
    _staging_device,
    gather_tensors,
    get_world_size,
    is_initialized,
    shard_slice,
)


def shard_rows(full: torch.Tensor) -> Tuple[torch.Tensor, int]:
    """This rank's contiguous row-shard of a replicated tensor and its
    global row offset."""
    s = shard_slice(full.shape[0])
    return full[s].contiguous(), s.start


def fold_rowmin_partials(dists, idxs) -> Tuple[torch.Tensor, torch.Tensor]:
    """Rank-ordered strict-less fold of per-shard (min, GLOBAL argmin)
    partials. With shards holding ascending global row ranges, strict-less
    in rank order preserves the single-device lowest-index tie rule —
    the determinism property tests/test_sharded.py pins with hypothesis.
    """
    best_d, best_i = dists[0], idxs[0]
    for r in range(1, len(dists)):
        take = dists[r] < best_d
        best_d = torch.where(take, dists[r], best_d)
        best_i = torch.where(take, idxs[r], best_i)
    return best_d, best_i


def sharded_rowmin_l2(
    test: torch.Tensor,
    train_shard: torch.Tensor,
    shard_offset: int,
    train_shard_norm: torch.Tensor = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Global (min L2 distance, global argmin) with the train rows sharded.

    Partials merge with :func:`fold_rowmin_partials`. A rank whose shard is
    empty (class smaller than the world size) contributes +inf partials and
    never wins the merge.
    """
    if train_shard.shape[0] == 0:
        d = torch.full(
            (test.shape[0],), float("inf"), dtype=test.dtype, device=test.device
        )
        i = torch.zeros(test.shape[0], dtype=torch.int64, device=test.device)
    else:
        d, i = ops.rowmin_l2(test, train_shard, train_shard_norm)
        i = i + shard_offset
    if not is_initialized():
        return d, i
    return fold_rowmin_partials(gather_tensors(d), gather_tensors(i))


def sharded_kde_logsumexp(
    test_w: torch.Tensor, train_shard_w: torch.Tensor
) -> torch.Tensor:
    """Global logsumexp_i(-0.5 ||t - x_i||^2) with train rows sharded.

    Per-rank partial LSEs merge with a rank-ordered streaming logsumexp —
    the partial-reduction pattern this workload has in place of
    ring-attention. Empty shards contribute -inf partials (the identity)."""
    if train_shard_w.shape[0] == 0:
        part = torch.full(
            (test_w.shape[0],), float("-inf"),
            dtype=test_w.dtype, device=test_w.device,
        )
    else:
        part = ops.kde_logsumexp(test_w, train_shard_w)
    if not is_initialized():
        return part
    stacked = torch.stack(gather_tensors(part))  # [world, m]
    return torch.logsumexp(stacked, dim=0)


def allreduce_bitmap_or(words: torch.Tensor) -> torch.Tensor:
    """Bitwise-OR all-reduce of packed int64 coverage-bitmap words, in place.

    The coverage-sharding collective (BASELINE config 4): each rank holds the
    profile rows of its test-input shard in a full-size [N, W] matrix (zeros
    outside its shard) and the OR union reassembles the full profile matrix
    identically on every rank. RCCL, like NCCL, exposes no BOR reduction op,
    so on the nccl backend the OR is an all-gather over xGMI followed by a
    deterministic local fold — these messages are KB-to-MB, latency-bound on
    7x153 GB/s xGMI either way. gloo carries BOR natively.
    """
    if not is_initialized():
        return words
    if dist.get_backend() == "gloo":
        staged, home = _staging_device(words)
        dist.all_reduce(staged, op=dist.ReduceOp.BOR)
        if home is not None:
            words.copy_(staged.to(home))
        return words
    parts = gather_tensors(words)
    world = get_world_size()
    acc = parts[0]
    for r in range(1, world):
        acc = torch.bitwise_or(acc, parts[r])
    words.copy_(acc)
    return words


def _allreduce_inplace(t: torch.Tensor, op) -> torch.Tensor:
    staged, home = _staging_device(t)
    dist.all_reduce(staged, op=op)
    if home is not None:
        t.copy_(staged.to(home))
    return t


def allreduce_minmax(mins: torch.Tensor, maxs: torch.Tensor):
    """Cross-rank elementwise min/max of aggregate statistics (K18)."""
    if is_initialized():
        _allreduce_inplace(mins, dist.ReduceOp.MIN)
        _allreduce_inplace(maxs, dist.ReduceOp.MAX)
    return mins, maxs


def allreduce_min(t: torch.Tensor) -> torch.Tensor:
    if is_initialized():
        _allreduce_inplace(t, dist.ReduceOp.MIN)
    return t


def allreduce_max(t: torch.Tensor) -> torch.Tensor:
    if is_initialized():
        _allreduce_inplace(t, dist.ReduceOp.MAX)
    return t


def allreduce_welford(count: float, mean: torch.Tensor, m2: torch.Tensor):
    """Chan-merge Welford partials across ranks (deterministic rank order)."""
    if not is_initialized():
        return count, mean, m2
    world = get_world_size()
    dev = mean.device
    counts = gather_tensors(
        torch.tensor([count], dtype=torch.float64, device=dev)
    )
    means = gather_tensors(mean)
    m2s = gather_tensors(m2)
    tot_c, tot_mean, tot_m2 = float(counts[0].item()), means[0], m2s[0]
    for r in range(1, world):
        c = float(counts[r].item())
        if c == 0:
            continue
        delta = means[r] - tot_mean
        new_c = tot_c + c
        tot_mean = tot_mean + delta * (c / new_c)
        tot_m2 = tot_m2 + m2s[r] + delta * delta * (tot_c * c / new_c)
        tot_c = new_c
    return tot_c, tot_mean, tot_m2
