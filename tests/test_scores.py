import numpy as np
import pytest
import torch

from simple_tip_amd import ops


def test_deep_gini_hand_values():
    probs = torch.tensor([[0.5, 0.5], [1.0, 0.0], [0.25, 0.75]])
    u = ops.softmax_uncertainties(probs)
    assert u["deep_gini"].tolist() == pytest.approx([0.5, 0.0, 1 - 0.0625 - 0.5625])


def test_softmax_and_pcs_signs():
    probs = torch.tensor([[0.7, 0.2, 0.1]])
    u = ops.softmax_uncertainties(probs)
    # confidence-type scores are negated (uwiz as_confidence=False)
    assert u["softmax"].item() == pytest.approx(-0.7)
    assert u["pcs"].item() == pytest.approx(-(0.7 - 0.2))


def test_entropy():
    probs = torch.tensor([[0.5, 0.5], [1.0, 0.0]])
    u = ops.softmax_uncertainties(probs)
    assert u["softmax_entropy"][0].item() == pytest.approx(np.log(2), rel=1e-5)
    assert u["softmax_entropy"][1].item() == pytest.approx(0.0, abs=1e-7)


def test_uncertainty_ordering_consistency():
    # a confident and an unconfident prediction: every score must rank the
    # unconfident one as more uncertain
    probs = torch.tensor([[0.98, 0.01, 0.01], [0.4, 0.35, 0.25]])
    u = ops.softmax_uncertainties(probs)
    for name, vals in u.items():
        assert vals[1] > vals[0], name


def test_variation_ratio():
    # 3 samples, 4 inputs
    preds = torch.tensor(
        [
            [0, 1, 2, 1],
            [0, 1, 0, 1],
            [0, 2, 0, 0],
        ]
    )
    mode, vr = ops.variation_ratio(preds, num_classes=3)
    assert mode.tolist() == [0, 1, 0, 1]
    assert vr.tolist() == pytest.approx([0.0, 1 / 3, 1 / 3, 1 / 3])
