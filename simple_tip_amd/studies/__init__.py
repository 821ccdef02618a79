"""Case studies: MNIST, Fashion-MNIST, CIFAR-10, IMDB (+ the ResNet-20
benchmark flagship). Data is synthetic (deterministic, class-structured) —
this environment has no dataset downloads; shapes/sizes match the reference
datasets (BASELINE.md row 'Dataset sizes')."""

from .registry import get_case_study, STUDIES

__all__ = ["get_case_study", "STUDIES"]
