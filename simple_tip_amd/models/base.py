"""Base class for tappable sequential models.

The reference builds a second "transparent" keras model that re-emits chosen
layer outputs (handler_model.py:193-206). In torch we simply return the
intermediate tensors from a single forward pass — the activations never
leave the device, which is the K15 "AT extraction fused into the forward
pass" hot path of the MI355X design (no host round-trip, no second model).
"""

from typing import List, Optional, Sequence, Tuple

import torch
import torch.nn as nn


class TapModel(nn.Module):
    """A model made of an indexed ``nn.ModuleList`` of layer stages.

    ``forward`` runs the stages in order and returns the final LOGITS
    (softmax is applied by consumers where probabilities are needed — the
    reference's keras models emit softmax directly, argmax semantics are
    identical).
    """

    #: subclasses set: number of classes
    num_classes: int = 0

    def __init__(self):
        super().__init__()
        self.layers = nn.ModuleList()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        for layer in self.layers:
            x = layer(x)
        return x

    @torch.no_grad()
    def forward_taps(
        self,
        x: torch.Tensor,
        tap_ids: Optional[Sequence[int]],
        include_output: bool = True,
    ) -> Tuple[List[torch.Tensor], torch.Tensor]:
        """One forward pass, returning ([tapped layer outputs], logits)."""
        taps = []
        wanted = set(tap_ids or [])
        for i, layer in enumerate(self.layers):
            x = layer(x)
            if i in wanted:
                taps.append(x)
        if not include_output and wanted:
            pass
        return taps, x

    def has_dropout(self) -> bool:
        """Whether MC-dropout sampling (variation ratio) is applicable."""
        return any(isinstance(m, nn.Dropout) for m in self.modules())
