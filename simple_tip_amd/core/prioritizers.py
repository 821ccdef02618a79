"""Prioritization orders: Coverage-Total (CTM) and Coverage-Additional (CAM).

API parity with reference src/core/prioritizers.py:7-59 (generators yielding
indices), implemented over packed bitmaps so the CAM greedy set-cover loop can
run device-resident on MI355X (ops.cam_order)."""

from typing import Generator, Union

import numpy as np
import torch

from .. import ops
from .bitmap import BitProfile


def _as_tensor(x) -> torch.Tensor:
    if isinstance(x, torch.Tensor):
        return x
    return torch.as_tensor(np.asarray(x))


def ctm(scores) -> Generator[int, None, None]:
    """Indices by decreasing score (Coverage-Total Method)."""
    scores = _as_tensor(scores)
    assert scores.dim() == 1
    for x in ops.ctm_order(scores).cpu().tolist():
        yield x


def cam(scores, profiles: Union[BitProfile, torch.Tensor, np.ndarray]) -> Generator[int, None, None]:
    """Indices by greedily maximising added coverage (Coverage-Additional).

    ``profiles`` may be a :class:`BitProfile` or a bool matrix (higher-rank
    inputs are flattened per sample, like the reference).
    """
    scores = _as_tensor(scores).float()
    if not isinstance(profiles, BitProfile):
        p = _as_tensor(profiles)
        p = p.reshape(p.shape[0], -1).bool()
        profiles = BitProfile.from_bool(p)
    order = ops.cam_order(scores, profiles.words, profiles.nbits)
    for x in order.cpu().tolist():
        yield x
