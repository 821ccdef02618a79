"""Neuron-coverage metrics: NAC, KMNC, NBC, SNAC, TKNC.

Capability parity with reference src/core/neuron_coverage.py:31-167, with two
MI355X-native changes:
- profiles are packed bitmaps (:class:`BitProfile`) produced by fused HIP
  kernels on device (bool numpy arrays in the reference);
- the score is the profile popcount, computed in the same pass
  (reference ``sum_score``, neuron_coverage.py:8-22).

Every metric takes a list of per-layer activation tensors [N, ...] and
returns ``(scores, BitProfile)``.
"""

import abc
from typing import List, Tuple

import torch

from .. import ops
from .bitmap import BitProfile


def flatten_layers(layers: List[torch.Tensor]) -> torch.Tensor:
    """Flatten each layer to [N, K_l] and concatenate along features."""
    flat = [l.reshape(l.shape[0], -1) for l in layers]
    return torch.cat(flat, dim=1)


def sum_score(profile: BitProfile) -> torch.Tensor:
    """Number of covered profile sections per sample."""
    return profile.popcount()


class CoverageMethod(abc.ABC):
    """Base class for coverage criteria (fit in __init__, profile in call)."""

    @abc.abstractmethod
    def __call__(
        self, activations: List[torch.Tensor]
    ) -> Tuple[torch.Tensor, BitProfile]:
        """Per-sample (scores, packed profiles) for a batch of activations."""


class NAC(CoverageMethod):
    """Neuron-Activation Coverage: activation > threshold."""

    def __init__(self, cov_threshold: float):
        self.cov_threshold = float(cov_threshold)

    def __call__(self, activations):
        acts = flatten_layers(activations)
        words = ops.nac_profile(acts, self.cov_threshold)
        prof = BitProfile(words, acts.shape[1])
        return sum_score(prof), prof


class KMNC(CoverageMethod):
    """K-Multisection Neuron Coverage over train min/max ranges."""

    def __init__(self, mins: List[torch.Tensor], maxs: List[torch.Tensor], sections: int):
        self.sections = int(sections)
        self.mins = flatten_layers([m.unsqueeze(0) for m in mins]).squeeze(0)
        self.maxs = flatten_layers([m.unsqueeze(0) for m in maxs]).squeeze(0)

    def __call__(self, activations):
        acts = flatten_layers(activations)
        words = ops.kmnc_profile(
            acts, self.mins.to(acts.device), self.maxs.to(acts.device), self.sections
        )
        prof = BitProfile(words, acts.shape[1] * self.sections)
        return sum_score(prof), prof


class NBC(CoverageMethod):
    """Neuron Boundary Coverage: a <= min - s*std or a >= max + s*std."""

    def __init__(self, mins, maxs, stds, scaler: float):
        min_arr = flatten_layers([m.unsqueeze(0) for m in mins]).squeeze(0)
        max_arr = flatten_layers([m.unsqueeze(0) for m in maxs]).squeeze(0)
        std_arr = flatten_layers([m.unsqueeze(0) for m in stds]).squeeze(0)
        self.min_boundaries = min_arr - scaler * std_arr
        self.max_boundaries = max_arr + scaler * std_arr

    def __call__(self, activations):
        acts = flatten_layers(activations)
        words = ops.nbc_profile(
            acts,
            self.min_boundaries.to(acts.device),
            self.max_boundaries.to(acts.device),
        )
        prof = BitProfile(words, acts.shape[1] * 2)
        return sum_score(prof), prof


class SNAC(CoverageMethod):
    """Strong Neuron Activation Coverage: a >= max + s*std."""

    def __init__(self, maxs, stds, scaler: float):
        max_arr = flatten_layers([m.unsqueeze(0) for m in maxs]).squeeze(0)
        std_arr = flatten_layers([m.unsqueeze(0) for m in stds]).squeeze(0)
        self.max_boundaries = max_arr + scaler * std_arr

    def __call__(self, activations):
        acts = flatten_layers(activations)
        words = ops.snac_profile(acts, self.max_boundaries.to(acts.device))
        prof = BitProfile(words, acts.shape[1])
        return sum_score(prof), prof


class TKNC(CoverageMethod):
    """Top-k Neuron Coverage: per layer, the k most-active neurons."""

    def __init__(self, top_neurons: int):
        self.top_neurons = int(top_neurons)

    def __call__(self, activations):
        layers = [l.reshape(l.shape[0], -1) for l in activations]
        words = ops.tknc_profile(layers, self.top_neurons)
        nbits = sum(l.shape[1] for l in layers)
        prof = BitProfile(words, nbits)
        return sum_score(prof), prof
