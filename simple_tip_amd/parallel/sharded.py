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
from .dist import get_rank, get_world_size, is_initialized, shard_slice


def shard_rows(full: torch.Tensor) -> Tuple[torch.Tensor, int]:
    """This rank's contiguous row-shard of a replicated tensor and its
    global row offset."""
    s = shard_slice(full.shape[0])
    return full[s].contiguous(), s.start


def sharded_rowmin_l2(
    test: torch.Tensor,
    train_shard: torch.Tensor,
    shard_offset: int,
    train_shard_norm: torch.Tensor = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Global (min L2 distance, global argmin) with the train rows sharded.

    Partials merge in rank order with strict-less comparison, preserving the
    lowest-global-index tie rule of the single-device path. A rank whose
    shard is empty (class smaller than the world size) contributes +inf
    partials and never wins the merge.
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
    world = get_world_size()
    dg = [torch.empty_like(d) for _ in range(world)]
    ig = [torch.empty_like(i) for _ in range(world)]
    dist.all_gather(dg, d.contiguous())
    dist.all_gather(ig, i.contiguous())
    best_d, best_i = dg[0], ig[0]
    for r in range(1, world):
        # ranks hold ascending global offsets: strict less keeps lowest idx
        take = dg[r] < best_d
        best_d = torch.where(take, dg[r], best_d)
        best_i = torch.where(take, ig[r], best_i)
    return best_d, best_i


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
    world = get_world_size()
    parts = [torch.empty_like(part) for _ in range(world)]
    dist.all_gather(parts, part.contiguous())
    stacked = torch.stack(parts)  # [world, m]
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
        dist.all_reduce(words, op=dist.ReduceOp.BOR)
        return words
    world = get_world_size()
    parts = [torch.empty_like(words) for _ in range(world)]
    dist.all_gather(parts, words.contiguous())
    acc = parts[0]
    for r in range(1, world):
        acc = torch.bitwise_or(acc, parts[r])
    words.copy_(acc)
    return words


def allreduce_minmax(mins: torch.Tensor, maxs: torch.Tensor):
    """Cross-rank elementwise min/max of aggregate statistics (K18)."""
    if is_initialized():
        dist.all_reduce(mins, op=dist.ReduceOp.MIN)
        dist.all_reduce(maxs, op=dist.ReduceOp.MAX)
    return mins, maxs


def allreduce_min(t: torch.Tensor) -> torch.Tensor:
    if is_initialized():
        dist.all_reduce(t, op=dist.ReduceOp.MIN)
    return t


def allreduce_max(t: torch.Tensor) -> torch.Tensor:
    if is_initialized():
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return t


def allreduce_welford(count: float, mean: torch.Tensor, m2: torch.Tensor):
    """Chan-merge Welford partials across ranks (deterministic rank order)."""
    if not is_initialized():
        return count, mean, m2
    world = get_world_size()
    dev = mean.device
    counts = [torch.zeros(1, dtype=torch.float64, device=dev) for _ in range(world)]
    means = [torch.empty_like(mean) for _ in range(world)]
    m2s = [torch.empty_like(m2) for _ in range(world)]
    dist.all_gather(counts, torch.tensor([count], dtype=torch.float64, device=dev))
    dist.all_gather(means, mean.contiguous())
    dist.all_gather(m2s, m2.contiguous())
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
