"""Streaming per-neuron min/max + Welford variance over train activations.

Capability parity with reference src/dnn_test_prio/aggregate_statistics.py
(timed min/max/Welford buckets for the coverage metrics' time debits), in
torch so the reductions run on device during the train-set AT pass (K18).
Cross-GPU combination (Chan merge of Welford partials, min/max all-reduce)
lives in parallel/dist.py.
"""

from typing import List, Tuple

import torch

from ..core.timer import DeviceTimer

AggStats = Tuple[List[torch.Tensor], List[torch.Tensor], List[torch.Tensor]]


class WelfordState:
    """Per-neuron running (count, mean, M2) with batch (Chan) updates."""

    def __init__(self, shape, device, dtype=torch.float64):
        self.count = 0.0
        self.mean = torch.zeros(shape, device=device, dtype=dtype)
        self.m2 = torch.zeros(shape, device=device, dtype=dtype)

    def add_batch(self, batch: torch.Tensor):
        b = batch.shape[0]
        if b == 0:
            return
        bx = batch.to(self.mean.dtype)
        bmean = bx.mean(dim=0)
        bm2 = ((bx - bmean) ** 2).sum(dim=0)
        if self.count == 0:
            self.mean, self.m2, self.count = bmean, bm2, float(b)
            return
        delta = bmean - self.mean
        tot = self.count + b
        self.mean = self.mean + delta * (b / tot)
        self.m2 = self.m2 + bm2 + delta * delta * (self.count * b / tot)
        self.count = tot

    def var_sample(self) -> torch.Tensor:
        """Sample variance (ddof=1), matching the welford package's var_s."""
        if self.count < 2:
            return torch.full_like(self.m2, float("nan"))
        return self.m2 / (self.count - 1)

    def merge(self, other: "WelfordState"):
        if other.count == 0:
            return
        if self.count == 0:
            self.count, self.mean, self.m2 = other.count, other.mean, other.m2
            return
        delta = other.mean - self.mean
        tot = self.count + other.count
        self.mean = self.mean + delta * (other.count / tot)
        self.m2 = self.m2 + other.m2 + delta * delta * (self.count * other.count / tot)
        self.count = tot


class AggregateStatisticsCollector:
    """Timed online min/max/std of equally shaped per-layer activations."""

    def __init__(self):
        self.initialized = False
        self.done = False
        self.mins: List[torch.Tensor] = []
        self.maxs: List[torch.Tensor] = []
        self.welfords: List[WelfordState] = []
        self.min_timer = DeviceTimer()
        self.max_timer = DeviceTimer()
        self.welford_timer = DeviceTimer()

    def track(self, badge: List[torch.Tensor]) -> None:
        """Fold the next batch of per-layer activations [B, ...] in."""
        if self.done:
            raise RuntimeError("`get` has been called; stats are frozen.")
        flat = [b.reshape(b.shape[0], -1) for b in badge]
        if not self.initialized:
            for layer in flat:
                with self.min_timer:
                    self.mins.append(layer.min(dim=0).values.clone())
                with self.max_timer:
                    self.maxs.append(layer.max(dim=0).values.clone())
                with self.welford_timer:
                    self.welfords.append(
                        WelfordState(layer.shape[1], layer.device)
                    )
            self.initialized = True
        with self.min_timer:
            for i, layer in enumerate(flat):
                torch.minimum(self.mins[i], layer.min(dim=0).values, out=self.mins[i])
        with self.max_timer:
            for i, layer in enumerate(flat):
                torch.maximum(self.maxs[i], layer.max(dim=0).values, out=self.maxs[i])
        with self.welford_timer:
            for i, layer in enumerate(flat):
                self.welfords[i].add_batch(layer)

    def cross_rank_merge(self) -> None:
        """Combine per-rank partial statistics across an initialised
        torch.distributed group (K18 cross-GPU combine): elementwise min/max
        all-reduce plus a rank-ordered Chan merge of the Welford partials.
        Call once, after tracking this rank's train shard, before `get`."""
        from ..parallel import sharded as shd
        from ..parallel.dist import is_initialized

        if not is_initialized():
            return
        with self.min_timer:
            for m in self.mins:
                shd.allreduce_min(m)
        with self.max_timer:
            for m in self.maxs:
                shd.allreduce_max(m)
        with self.welford_timer:
            for w in self.welfords:
                w.count, w.mean, w.m2 = shd.allreduce_welford(
                    w.count, w.mean, w.m2
                )

    def get(self) -> AggStats:
        """(mins, maxs, stds) per layer (flattened per-neuron vectors)."""
        self.done = True
        with self.welford_timer:
            stds = [
                torch.sqrt(w.var_sample()).to(self.mins[i].dtype)
                for i, w in enumerate(self.welfords)
            ]
        return self.mins, self.maxs, stds
