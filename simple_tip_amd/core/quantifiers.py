"""Uncertainty-quantifier registry (uncertainty-wizard replacement).

The reference registers quantifiers by alias in uwiz's registry (reference
deepgini.py:12-40, handler_model.py:16-20). This is the native equivalent:
each quantifier maps softmax outputs [N, C] to (point predictions,
per-input quantification), declares whether higher means confidence or
uncertainty, and is resolvable by alias. The registry is what
``BaseModel.get_pred_and_uncertainty`` conceptually evaluates — on device
all point quantifiers run as one fused kernel pass (ops.softmax_uncertainties).
"""

from typing import Callable, Dict, List, Tuple

import torch

from .. import ops


class Quantifier:
    """A softmax-output quantifier."""

    aliases: List[str] = []
    is_confidence = False
    takes_samples = False

    def calculate(self, nn_outputs: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        raise NotImplementedError

    def as_uncertainty(self, nn_outputs: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """(predictions, quantification with higher = more uncertain)."""
        pred, q = self.calculate(nn_outputs)
        return pred, (-q if self.is_confidence else q)


def _pred(probs):
    return probs.argmax(dim=1)


class MaxSoftmax(Quantifier):
    aliases = ["softmax", "max_softmax", "MaxSoftmax", "SM"]
    is_confidence = True

    def calculate(self, probs):
        return _pred(probs), probs.max(dim=1).values


class PredictionConfidenceScore(Quantifier):
    aliases = ["pcs", "PCS", "prediction_confidence_score"]
    is_confidence = True

    def calculate(self, probs):
        top2 = torch.topk(probs, k=min(2, probs.shape[1]), dim=1).values
        p2 = top2[:, 1] if probs.shape[1] > 1 else torch.zeros_like(top2[:, 0])
        return _pred(probs), top2[:, 0] - p2


class SoftmaxEntropy(Quantifier):
    aliases = ["softmax_entropy", "SE", "entropy"]
    is_confidence = False

    def calculate(self, probs):
        logp = torch.where(probs > 0, torch.log(probs), torch.zeros_like(probs))
        return _pred(probs), -(probs * logp).sum(dim=1)


class DeepGini(Quantifier):
    """1 - sum(softmax^2) (reference deepgini.py:32-35)."""

    aliases = ["custom::deep_gini", "deep_gini", "DeepGini"]
    is_confidence = False

    def calculate(self, probs):
        return _pred(probs), 1.0 - (probs * probs).sum(dim=1)


class VariationRatio(Quantifier):
    """MC-dropout variation ratio over sampled class predictions [S, N]."""

    aliases = ["VR", "var_ratio", "variation_ratio"]
    is_confidence = False
    takes_samples = True

    def calculate(self, sample_preds, num_classes=None):
        nc = num_classes or int(sample_preds.max().item()) + 1
        mode, vr = ops.variation_ratio(sample_preds, nc)
        return mode, vr


class QuantifierRegistry:
    """Alias -> quantifier lookup; extensible like the uwiz registry."""

    _by_alias: Dict[str, Quantifier] = {}

    @classmethod
    def register(cls, q: Quantifier):
        for a in q.aliases:
            if a in cls._by_alias:
                raise ValueError(f"alias already registered: {a}")
            cls._by_alias[a] = q

    @classmethod
    def find(cls, alias: str) -> Quantifier:
        return cls._by_alias[alias]


for _q in (MaxSoftmax(), PredictionConfidenceScore(), SoftmaxEntropy(),
           DeepGini(), VariationRatio()):
    QuantifierRegistry.register(_q)
