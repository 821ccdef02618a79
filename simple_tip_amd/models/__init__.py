"""PyTorch-ROCm model zoo with activation-trace taps.

All models expose an indexed layer list so tap indices stay compatible with
the reference's keras ``model.layers`` indexing (reference
handler_model.py:193-206)."""

from .base import TapModel
from .cnn import Cifar10CNN, MnistCNN, ResNet20
from .transformer import ImdbTransformer

__all__ = ["TapModel", "MnistCNN", "Cifar10CNN", "ResNet20", "ImdbTransformer"]
