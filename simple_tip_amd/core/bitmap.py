"""Packed coverage-profile bitmaps.

The reference keeps coverage profiles as huge bool numpy arrays
(src/core/neuron_coverage.py) and spills them to disk
(handler_coverage.py:165-187). On MI355X we keep them resident in HBM as
64-bit words (64x smaller than bool bytes, popcount-friendly for the CAM
kernel); this class is the common container for both the CPU fallback and
the HIP kernels.
"""

from dataclasses import dataclass
from typing import List

import torch

from .. import ops


@dataclass
class BitProfile:
    """A packed [N, ceil(nbits/64)] int64 coverage-profile matrix."""

    words: torch.Tensor
    nbits: int

    @property
    def n(self) -> int:
        return self.words.shape[0]

    @staticmethod
    def from_bool(profile: torch.Tensor) -> "BitProfile":
        """Pack a bool [N, K] tensor."""
        return BitProfile(ops.pack_bits(profile), profile.shape[1])

    def to_bool(self) -> torch.Tensor:
        """Unpack to a bool [N, nbits] tensor (CPU-side; for tests)."""
        return ops.unpack_bits(self.words, self.nbits)

    def popcount(self) -> torch.Tensor:
        """Per-row number of set bits (the reference's ``sum_score``)."""
        return ops.popcount_rows(self.words)

    @staticmethod
    def cat(profiles: List["BitProfile"]) -> "BitProfile":
        """Concatenate along the sample axis (all parts share nbits)."""
        nbits = profiles[0].nbits
        assert all(p.nbits == nbits for p in profiles)
        return BitProfile(torch.cat([p.words for p in profiles], dim=0), nbits)

    def to(self, device) -> "BitProfile":
        return BitProfile(self.words.to(device), self.nbits)
