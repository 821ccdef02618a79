"""Device-dispatching TIP ops.

Every op has two implementations:
- ``fallback``: pure torch/numpy (the CPU path and the GPU-test oracle)
- ``_tip_hip``: hand-written HIP/CDNA4 kernels (gfx950), built in-tree by
  ``setup.py`` / ``__graft_entry__.build()``.

Dispatch rule: CUDA tensors REQUIRE the HIP extension — on a GPU box a
missing/failed extension raises instead of silently falling back to eager
torch (set ``TIP_ALLOW_GPU_FALLBACK=1`` only for debugging). CPU tensors use
the fallback.
"""

import os
from typing import Dict, List, Tuple

import torch

from . import fallback

_EXT = None
_EXT_ERR = None


def _load_compiled():
    """Import the compiled _tip_hip extension module (raises ImportError)."""
    from . import _tip_hip  # built in-tree by setup.py build_ext --inplace

    return _tip_hip


def _load_ext():
    """Load the HIP op wrapper (returns None if the .so is unavailable)."""
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from . import hip_ops  # thin wrapper over the compiled _tip_hip .so

        _EXT = hip_ops
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = e
        _EXT = None
    return _EXT


def hip_available() -> bool:
    """True iff the HIP extension is importable."""
    return _load_ext() is not None


def _route(t: torch.Tensor):
    """Return the implementing module for a tensor's device."""
    if t.is_cuda:
        ext = _load_ext()
        if ext is None:
            if os.environ.get("TIP_ALLOW_GPU_FALLBACK") == "1":
                return fallback
            raise RuntimeError(
                "simple_tip_amd: tensor is on GPU but the HIP extension "
                "_tip_hip is not importable (build it with "
                "`python setup.py build_ext --inplace`); refusing to fall "
                f"back to eager torch. Import error: {_EXT_ERR!r}"
            )
        return ext
    return fallback


# --- bitmap ops -------------------------------------------------------------

def pack_bits(profile: torch.Tensor) -> torch.Tensor:
    return _route(profile).pack_bits(profile)


def unpack_bits(words: torch.Tensor, nbits: int) -> torch.Tensor:
    return fallback.unpack_bits(words, nbits)


def popcount_rows(words: torch.Tensor) -> torch.Tensor:
    return _route(words).popcount_rows(words)


def ctm_order(scores: torch.Tensor) -> torch.Tensor:
    return fallback.ctm_order(scores)


def cam_order(scores: torch.Tensor, words: torch.Tensor, nbits: int) -> torch.Tensor:
    return _route(words).cam_order(scores, words, nbits)


# --- pairwise-distance family ----------------------------------------------

def pairwise_sqdist(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    return _route(a).pairwise_sqdist(a.contiguous(), b.contiguous())


def rowmin_l2(
    a: torch.Tensor, b: torch.Tensor, bnorm: torch.Tensor = None
) -> Tuple[torch.Tensor, torch.Tensor]:
    return _route(a).rowmin_l2(a.contiguous(), b.contiguous(), bnorm)


def kde_logsumexp(test_w: torch.Tensor, train_w: torch.Tensor) -> torch.Tensor:
    return _route(test_w).kde_logsumexp(test_w.contiguous(), train_w.contiguous())


# --- score / profile ops ----------------------------------------------------

def softmax_uncertainties(probs: torch.Tensor) -> Dict[str, torch.Tensor]:
    return _route(probs).softmax_uncertainties(probs.contiguous())


def variation_ratio(sample_preds: torch.Tensor, num_classes: int):
    return fallback.variation_ratio(sample_preds, num_classes)


def nac_profile(acts: torch.Tensor, threshold: float) -> torch.Tensor:
    return _route(acts).nac_profile(acts.contiguous(), threshold)


def snac_profile(acts: torch.Tensor, max_bound: torch.Tensor) -> torch.Tensor:
    return _route(acts).snac_profile(acts.contiguous(), max_bound.contiguous())


def nbc_profile(acts: torch.Tensor, min_bound: torch.Tensor, max_bound: torch.Tensor) -> torch.Tensor:
    return _route(acts).nbc_profile(
        acts.contiguous(), min_bound.contiguous(), max_bound.contiguous()
    )


def kmnc_profile(acts: torch.Tensor, mins: torch.Tensor, maxs: torch.Tensor, sections: int) -> torch.Tensor:
    return _route(acts).kmnc_profile(
        acts.contiguous(), mins.contiguous(), maxs.contiguous(), sections
    )


def tknc_profile(layer_acts: List[torch.Tensor], k: int) -> torch.Tensor:
    return _route(layer_acts[0]).tknc_profile(
        [l.contiguous() for l in layer_acts], k
    )


def bucketize_profile(values: torch.Tensor, thresholds: torch.Tensor) -> torch.Tensor:
    return _route(values).bucketize_profile(values, thresholds)
