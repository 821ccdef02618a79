"""GPU implementations of the TIP ops, backed by the _tip_hip extension.

Presents the same API as ops/fallback.py; ops/__init__.py routes CUDA
tensors here. Import fails (propagated by the dispatcher as a loud error)
when the extension .so is missing."""

from typing import Dict, List, Tuple

import torch

from . import _load_compiled

_ext = _load_compiled()

# profile kernel modes (must match coverage.hip ProfMode)
_PROF_NAC, _PROF_SNAC, _PROF_NBC, _PROF_KMNC, _PROF_PACK = range(5)


def pack_bits(profile: torch.Tensor) -> torch.Tensor:
    return _ext.pack_bits(profile.contiguous())


def unpack_bits(words: torch.Tensor, nbits: int) -> torch.Tensor:
    from . import fallback

    return fallback.unpack_bits(words, nbits)


def popcount_rows(words: torch.Tensor) -> torch.Tensor:
    return _ext.popcount_rows(words.contiguous())


def ctm_order(scores: torch.Tensor) -> torch.Tensor:
    return torch.argsort(scores, descending=True, stable=True)


def cam_order(scores: torch.Tensor, words: torch.Tensor, nbits: int) -> torch.Tensor:
    picked = _ext.cam_greedy(words.contiguous(), nbits).to(scores.device)
    n = scores.shape[0]
    mask = torch.ones(n, dtype=torch.bool, device=scores.device)
    if picked.numel():
        mask[picked] = False
    left = torch.nonzero(mask, as_tuple=True)[0]
    if left.numel():
        order_left = left[
            torch.argsort(scores[left].float(), descending=True, stable=True)
        ]
        return torch.cat([picked, order_left])
    return picked


def pairwise_sqdist(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    return _ext.pairwise_sqdist(a.float(), b.float())


def rowmin_l2(
    a: torch.Tensor, b: torch.Tensor, bnorm: torch.Tensor = None
) -> Tuple[torch.Tensor, torch.Tensor]:
    d, i = _ext.rowmin_l2(a.float(), b.float(), bnorm)
    return d, i


def kde_logsumexp(test_w: torch.Tensor, train_w: torch.Tensor) -> torch.Tensor:
    return _ext.kde_logsumexp(test_w.float(), train_w.float())


def softmax_uncertainties(probs: torch.Tensor) -> Dict[str, torch.Tensor]:
    neg_max, neg_pcs, entropy, gini = _ext.softmax_scores(probs.float())
    return {
        "softmax": neg_max,
        "pcs": neg_pcs,
        "softmax_entropy": entropy,
        "deep_gini": gini,
    }


def variation_ratio(sample_preds: torch.Tensor, num_classes: int):
    from . import fallback

    return fallback.variation_ratio(sample_preds, num_classes)


def _empty(t):
    return torch.empty(0, dtype=torch.float32, device=t.device)


def nac_profile(acts: torch.Tensor, threshold: float) -> torch.Tensor:
    words, _ = _ext.profile(
        _PROF_NAC, acts.float(), _empty(acts), _empty(acts), float(threshold),
        1, acts.shape[1],
    )
    return words


def snac_profile(acts: torch.Tensor, max_bound: torch.Tensor) -> torch.Tensor:
    words, _ = _ext.profile(
        _PROF_SNAC, acts.float(), _empty(acts), max_bound.float(), 0.0, 1,
        acts.shape[1],
    )
    return words


def nbc_profile(acts, min_bound, max_bound) -> torch.Tensor:
    words, _ = _ext.profile(
        _PROF_NBC, acts.float(), min_bound.float(), max_bound.float(), 0.0, 2,
        acts.shape[1] * 2,
    )
    return words


def kmnc_profile(acts, mins, maxs, sections: int) -> torch.Tensor:
    words, _ = _ext.profile(
        _PROF_KMNC, acts.float(), mins.float(), maxs.float(), 0.0, sections,
        acts.shape[1] * sections,
    )
    return words


def tknc_profile(layer_acts: List[torch.Tensor], k: int) -> torch.Tensor:
    nbits = sum(l.shape[1] for l in layer_acts)
    n = layer_acts[0].shape[0]
    w = (nbits + 63) // 64
    words = torch.zeros(n, w, dtype=torch.int64, device=layer_acts[0].device)
    offset = 0
    for layer in layer_acts:
        _ext.tknc_layer(layer.float(), k, offset, words)
        offset += layer.shape[1]
    return words


def bucketize_profile(values: torch.Tensor, thresholds: torch.Tensor) -> torch.Tensor:
    return _ext.bucketize(
        values.double().contiguous(),
        thresholds.to(values.device).double().contiguous(),
        thresholds.shape[0] - 1,
    )
