"""KMeans (Lloyd, kmeans++ init) and silhouette score on torch tensors.

Replaces sklearn's KMeans/silhouette_score used by the reference's
``_KmeansDiscriminator`` (src/core/surprise.py:102-133). Both the Lloyd
assignment step and the silhouette score are pairwise-distance problems, so
on MI355X they run on the same MFMA pairwise-sqdist kernel as DSA/KDE
(ops.pairwise_sqdist / ops.rowmin_l2). Deterministic for a given seed.
"""

from typing import Tuple

import torch

from .. import ops


def _kmeanspp_init(x: torch.Tensor, k: int, gen: torch.Generator) -> torch.Tensor:
    n = x.shape[0]
    centers = torch.empty(k, x.shape[1], dtype=x.dtype, device=x.device)
    first = int(torch.randint(n, (1,), generator=gen).item())
    centers[0] = x[first]
    d2 = ops.pairwise_sqdist(x, centers[0:1]).squeeze(1).clamp_min_(0)
    for i in range(1, k):
        probs = d2 / d2.sum().clamp_min(1e-30)
        idx = int(torch.multinomial(probs.cpu(), 1, generator=gen).item())
        centers[i] = x[idx]
        nd = ops.pairwise_sqdist(x, centers[i : i + 1]).squeeze(1).clamp_min_(0)
        d2 = torch.minimum(d2, nd)
    return centers


def kmeans_fit(
    x: torch.Tensor,
    k: int,
    n_init: int = 10,
    max_iter: int = 300,
    seed: int = 0,
    tol: float = 1e-4,
) -> Tuple[torch.Tensor, torch.Tensor, float]:
    """Fit k centers; returns (centers, labels, inertia). Best of n_init."""
    gen = torch.Generator().manual_seed(seed)
    best = None
    for _ in range(n_init):
        centers = _kmeanspp_init(x, k, gen)
        for _ in range(max_iter):
            d, labels = ops.rowmin_l2(x, centers)
            new_centers = torch.zeros_like(centers)
            counts = torch.zeros(k, dtype=x.dtype, device=x.device)
            new_centers.index_add_(0, labels, x)
            counts.index_add_(0, labels, torch.ones_like(d))
            empty = counts == 0
            counts = counts.clamp_min(1.0)
            new_centers /= counts.unsqueeze(1)
            # keep empty clusters where they were (sklearn reseeds; rare)
            new_centers[empty] = centers[empty]
            shift = (new_centers - centers).pow(2).sum()
            centers = new_centers
            if float(shift) <= tol:
                break
        d, labels = ops.rowmin_l2(x, centers)
        inertia = float((d * d).sum())
        if best is None or inertia < best[2]:
            best = (centers, labels, inertia)
    return best


def kmeans_predict(x: torch.Tensor, centers: torch.Tensor) -> torch.Tensor:
    """Assign each row of x to its nearest center."""
    _, labels = ops.rowmin_l2(x, centers)
    return labels


def silhouette_score(x: torch.Tensor, labels: torch.Tensor, chunk: int = 2048) -> float:
    """Mean silhouette coefficient over all samples.

    s(i) = (b(i) - a(i)) / max(a(i), b(i)) with a = mean intra-cluster
    distance (excluding self), b = min mean distance to another cluster.
    Computed chunked so the N x N distance matrix never materialises.
    """
    labels = labels.to(x.device)
    k = int(labels.max().item()) + 1
    n = x.shape[0]
    onehot = torch.zeros(n, k, dtype=x.dtype, device=x.device)
    onehot[torch.arange(n, device=x.device), labels] = 1.0
    counts = onehot.sum(dim=0)  # [k]
    s_vals = torch.empty(n, dtype=x.dtype, device=x.device)
    for s0 in range(0, n, chunk):
        xc = x[s0 : s0 + chunk]
        d = ops.pairwise_sqdist(xc, x).clamp_min_(0).sqrt_()  # [c, N]
        sums = d @ onehot  # [c, k] sum of distances to each cluster
        own = labels[s0 : s0 + chunk]
        own_counts = counts[own]
        a = sums[torch.arange(xc.shape[0], device=x.device), own] / (
            own_counts - 1
        ).clamp_min(1.0)
        mean_to = sums / counts.clamp_min(1.0)
        # empty clusters (kmeans_fit can retain a center nobody claims) must
        # not offer b=0 to every sample — sklearn raises on such labelings;
        # masking them to +inf drops them from the min
        mean_to[:, counts == 0] = float("inf")
        mean_to[torch.arange(xc.shape[0], device=x.device), own] = float("inf")
        b = mean_to.min(dim=1).values
        s = (b - a) / torch.maximum(a, b)
        # singleton clusters have s = 0 by convention
        s = torch.where(own_counts > 1, s, torch.zeros_like(s))
        s_vals[s0 : s0 + chunk] = s
    return float(s_vals.mean())
