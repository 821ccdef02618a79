"""Pure torch/numpy reference implementations of every TIP hot-path op.

These serve two purposes:
1. The CPU execution path (this container has no GPU; CI runs here).
2. The numeric oracle the HIP/CDNA4 kernels are tested against
   (tests/test_gpu_*.py compare device results to these).

Bit layout contract (shared with the HIP kernels in simple_tip_amd/ops/hip):
profiles are packed little-endian into 64-bit words — bit ``j`` of word ``w``
of a row is profile column ``w*64 + j``. For multi-section metrics
(KMNC sections, NBC sides) the column index is ``neuron * S + s``, matching
the reference's row-major ``(N, K, S)`` flattening
(reference: src/core/neuron_coverage.py:82-94, 118-128).
"""

from typing import Dict, Tuple

import numpy as np
import torch

# ---------------------------------------------------------------------------
# Bitmap packing / popcount
# ---------------------------------------------------------------------------

_POPCOUNT_U8 = np.array(
    [bin(i).count("1") for i in range(256)], dtype=np.int64
)


def pack_bits(profile: torch.Tensor) -> torch.Tensor:
    """Pack a bool tensor [N, K] into int64 words [N, ceil(K/64)] (LSB-first)."""
    assert profile.dtype == torch.bool and profile.dim() == 2
    arr = profile.cpu().numpy()
    n, k = arr.shape
    w = (k + 63) // 64
    padded = np.zeros((n, w * 64), dtype=np.uint8)
    padded[:, :k] = arr
    # numpy packbits is MSB-first per byte; request little bit order to get
    # our LSB-first contract, then view the 8-byte groups as uint64.
    packed = np.packbits(padded, axis=1, bitorder="little")
    words = packed.view(np.uint64).astype(np.int64, copy=False)
    return torch.from_numpy(np.ascontiguousarray(words)).to(profile.device)


def unpack_bits(words: torch.Tensor, nbits: int) -> torch.Tensor:
    """Inverse of :func:`pack_bits` → bool tensor [N, nbits]."""
    arr = words.cpu().numpy().view(np.uint64).view(np.uint8)
    n = words.shape[0]
    bits = np.unpackbits(arr.reshape(n, -1), axis=1, bitorder="little")
    return torch.from_numpy(bits[:, :nbits].astype(bool)).to(words.device)


def popcount_rows(words: torch.Tensor) -> torch.Tensor:
    """Per-row popcount of packed profiles [N, W] → int64 [N]."""
    arr = words.cpu().numpy().view(np.uint8)
    counts = _POPCOUNT_U8[arr.reshape(words.shape[0], -1)].sum(axis=1)
    return torch.from_numpy(counts).to(words.device)


# ---------------------------------------------------------------------------
# Prioritization orders
# ---------------------------------------------------------------------------


def ctm_order(scores: torch.Tensor) -> torch.Tensor:
    """Coverage-Total Method: indices by descending score, ties by index.

    Reference: src/core/prioritizers.py:7-13 (np.argsort(-scores)).
    """
    assert scores.dim() == 1
    order = torch.argsort(scores, descending=True, stable=True)
    return order


def cam_order(scores: torch.Tensor, words: torch.Tensor, nbits: int) -> torch.Tensor:
    """Coverage-Additional Method over packed profiles.

    Greedy max-cover: repeatedly pick the row covering the most
    still-uncovered columns (first index on ties, like np.argmax), mark those
    columns covered, repeat until no row adds coverage; remaining rows follow
    by descending original score (stable). Semantics match reference
    src/core/prioritizers.py:16-59.
    """
    assert scores.dim() == 1 and words.dim() == 2
    n = scores.shape[0]
    w_np = words.cpu().numpy().view(np.uint64).copy()
    uncovered = np.full(w_np.shape[1], ~np.uint64(0), dtype=np.uint64)
    # zero the padding bits of the last word so `remaining` is exact
    tail = nbits % 64
    if tail:
        uncovered[-1] = np.uint64((1 << tail) - 1)
    num_coverable = _popcount_np(w_np & uncovered)
    remaining = nbits
    yielded = np.zeros(n, dtype=bool)
    order = []
    while True:
        nxt = int(np.argmax(num_coverable))
        newly = int(num_coverable[nxt])
        if newly == 0:
            break
        order.append(nxt)
        yielded[nxt] = True
        newly_mask = w_np[nxt] & uncovered
        num_coverable -= _popcount_np(w_np & newly_mask)
        uncovered &= ~newly_mask
        remaining -= newly
        if remaining == 0:
            break
    # Leftovers: descending original score, stable (ties by index).
    if not np.all(yielded):
        s = scores.cpu().numpy()
        left = np.where(~yielded)[0]
        left = left[np.argsort(-s[left], kind="stable")]
        order.extend(int(i) for i in left)
    return torch.tensor(order, dtype=torch.long, device=scores.device)


def _popcount_np(w: np.ndarray) -> np.ndarray:
    """Row-wise popcount of a 2D uint64 array."""
    assert w.ndim == 2
    return _POPCOUNT_U8[w.view(np.uint8).reshape(w.shape[0], -1)].sum(axis=1)


# ---------------------------------------------------------------------------
# Pairwise distances (DSA / KDE / Mahalanobis / kmeans core)
# ---------------------------------------------------------------------------


def pairwise_sqdist(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Exact squared L2 distances [N, M] between rows of a [N,D] and b [M,D]."""
    d = torch.cdist(
        a.unsqueeze(0), b.unsqueeze(0), compute_mode="donot_use_mm_for_euclid_dist"
    ).squeeze(0)
    return d * d


def rowmin_l2(
    a: torch.Tensor, b: torch.Tensor, bnorm: torch.Tensor = None, chunk: int = 4096
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-row (min L2 distance, argmin) from rows of a to rows of b.

    Ties resolve to the lowest b-index (np.argmin semantics).
    """
    mins = torch.empty(a.shape[0], dtype=a.dtype, device=a.device)
    args = torch.empty(a.shape[0], dtype=torch.long, device=a.device)
    for s in range(0, a.shape[0], chunk):
        d = torch.cdist(
            a[s : s + chunk].unsqueeze(0),
            b.unsqueeze(0),
            compute_mode="donot_use_mm_for_euclid_dist",
        ).squeeze(0)
        m, idx = d.min(dim=1)
        mins[s : s + chunk] = m
        args[s : s + chunk] = idx
    return mins, args


def kde_logsumexp(
    test_w: torch.Tensor, train_w: torch.Tensor, chunk: int = 2048
) -> torch.Tensor:
    """logsumexp_i(-0.5 * ||t - x_i||^2) per test row, over whitened coords.

    The Gaussian-KDE hot loop: with both sides whitened by the bandwidth
    Cholesky factor, the kernel sum is a pairwise-sqdist + logsumexp epilogue
    (reference equivalent: scipy gaussian_kde.evaluate via
    src/core/stable_kde.py:79-101, computed there in float64 without the
    log-domain stabilisation we add here).
    """
    out = torch.empty(test_w.shape[0], dtype=test_w.dtype, device=test_w.device)
    for s in range(0, test_w.shape[0], chunk):
        d2 = pairwise_sqdist(test_w[s : s + chunk], train_w)
        out[s : s + chunk] = torch.logsumexp(-0.5 * d2, dim=1)
    return out


# ---------------------------------------------------------------------------
# Softmax-family uncertainty scores
# ---------------------------------------------------------------------------


def softmax_uncertainties(probs: torch.Tensor) -> Dict[str, torch.Tensor]:
    """All point-prediction uncertainty scores from softmax outputs [N, C].

    Naming and sign conventions follow the reference artifacts
    (uncertainty-wizard quantifiers with ``as_confidence=False`` negate
    confidence-type scores; reference handler_model.py:23-86,133-139 and
    plotters/utils.py approach names):
      - ``softmax``          = -max(p)              (negated MaxSoftmax)
      - ``pcs``              = -(p_top1 - p_top2)   (negated PredConfidence)
      - ``softmax_entropy``  = -sum p log p         (natural log; 0 log 0 = 0)
      - ``deep_gini``        = 1 - sum p^2          (reference deepgini.py:32-35)
    """
    top2 = torch.topk(probs, k=min(2, probs.shape[1]), dim=1).values
    p1 = top2[:, 0]
    p2 = top2[:, 1] if probs.shape[1] > 1 else torch.zeros_like(p1)
    logp = torch.where(probs > 0, torch.log(probs), torch.zeros_like(probs))
    entropy = -(probs * logp).sum(dim=1)
    gini = 1.0 - (probs * probs).sum(dim=1)
    return {
        "softmax": -p1,
        "pcs": -(p1 - p2),
        "softmax_entropy": entropy,
        "deep_gini": gini,
    }


def variation_ratio(sample_preds: torch.Tensor, num_classes: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """MC-dropout variation ratio from sampled class predictions [S, N].

    Returns (mode prediction, 1 - mode_count/S). Ties resolve to the lowest
    class index.
    """
    s, n = sample_preds.shape
    onehot = torch.zeros(n, num_classes, dtype=torch.float32, device=sample_preds.device)
    onehot.scatter_add_(
        1,
        sample_preds.t().long(),
        torch.ones(n, s, dtype=torch.float32, device=sample_preds.device),
    )
    counts, mode = onehot.max(dim=1)
    vr = 1.0 - counts / float(s)
    return mode, vr


# ---------------------------------------------------------------------------
# Coverage profiling (bool → packed bitmaps)
# ---------------------------------------------------------------------------


def nac_profile(acts: torch.Tensor, threshold: float) -> torch.Tensor:
    """NAC: bit per neuron, set iff activation > threshold. Returns packed."""
    return pack_bits(acts > threshold)


def snac_profile(acts: torch.Tensor, max_bound: torch.Tensor) -> torch.Tensor:
    """SNAC: activation >= max + scaler*std (bound precomputed)."""
    return pack_bits(acts >= max_bound)


def nbc_profile(
    acts: torch.Tensor, min_bound: torch.Tensor, max_bound: torch.Tensor
) -> torch.Tensor:
    """NBC: two bits per neuron — a <= min_bound, a >= max_bound.

    Column layout: ``neuron*2 + side`` (side 0 = lower, 1 = upper).
    """
    lower = acts <= min_bound
    upper = acts >= max_bound
    prof = torch.stack([lower, upper], dim=2).reshape(acts.shape[0], -1)
    return pack_bits(prof)


def kmnc_profile(
    acts: torch.Tensor, mins: torch.Tensor, maxs: torch.Tensor, sections: int
) -> torch.Tensor:
    """KMNC: per-neuron k-section membership bits (column = neuron*S + s).

    Section s covers [min + jump*s, min + jump*(s+1)); values equal to max or
    outside [min, max) set no bit — matching reference
    neuron_coverage.py:82-94.
    """
    n, k = acts.shape
    jumps = (maxs - mins) / sections
    prof = torch.zeros(n, k, sections, dtype=torch.bool, device=acts.device)
    for s in range(sections):
        lo = mins + jumps * s
        hi = mins + jumps * (s + 1)
        prof[..., s] = (lo <= acts) & (acts < hi)
    return pack_bits(prof.reshape(n, -1))


def tknc_profile(layer_acts, k: int) -> torch.Tensor:
    """TKNC: per layer, bit set for each of the layer's top-k neurons.

    ``layer_acts`` is a list of [N, K_l] tensors; columns are the flattened
    concatenation of layers (reference neuron_coverage.py:155-167).
    """
    parts = []
    for layer in layer_acts:
        flat = layer.reshape(layer.shape[0], -1)
        kk = min(k, flat.shape[1])
        idx = torch.topk(flat, k=kk, dim=1).indices
        prof = torch.zeros_like(flat, dtype=torch.bool)
        prof.scatter_(1, idx, True)
        parts.append(prof)
    return pack_bits(torch.cat(parts, dim=1))


def bucketize_profile(values: torch.Tensor, thresholds: torch.Tensor) -> torch.Tensor:
    """Surprise-coverage binning: bit s set iff thr[s] <= v < thr[s+1].

    ``thresholds`` has S+1 entries; values outside [thr[0], thr[S]) (including
    v == thr[S] exactly) set no bit — matching reference surprise.py:186-209.
    """
    s = thresholds.shape[0] - 1
    idx = torch.searchsorted(thresholds, values.to(thresholds.dtype), right=True) - 1
    valid = (idx >= 0) & (idx < s)
    # note: searchsorted(right=True) maps v == thr[i] to bucket i, and
    # v == thr[S] to S (invalid) — the reference's half-open intervals.
    prof = torch.zeros(values.shape[0], s, dtype=torch.bool)
    rows = torch.nonzero(valid, as_tuple=True)[0]
    prof[rows, idx[valid]] = True
    return pack_bits(prof).to(values.device)
