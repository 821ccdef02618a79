"""Quality-signal pins: trained models must have a
real-but-small nominal error rate (nominal APFD well-defined, not NaN), and
the deep-gini ordering on a trained model's NOMINAL batch must beat a random
ordering — the reference's headline Table-1 qualitative claim
(reference eval_apfd_table.py:111-131)."""

import numpy as np
import pytest
import torch

from simple_tip_amd import ops
from simple_tip_amd.core.apfd import apfd_from_order
from simple_tip_amd.models import MnistCNN
from simple_tip_amd.studies.base import train_classifier
from simple_tip_amd.studies.synthetic import synthetic_images


@pytest.fixture(scope="module")
def trained_model_and_nominal():
    torch.manual_seed(0)
    tx, ty = synthetic_images("qsig", "train", 3000, (1, 28, 28), 10)
    nx, ny = synthetic_images("qsig", "test", 1024, (1, 28, 28), 10)
    model = train_classifier(
        MnistCNN(), tx, ty, epochs=4, batch_size=128,
        device=torch.device("cpu"), seed=0,
    )
    model.eval()
    with torch.no_grad():
        logits = model(torch.from_numpy(nx))
        probs = torch.softmax(logits, dim=1)
    pred = probs.argmax(dim=1).numpy()
    return probs, pred, ny


def test_nominal_error_rate_is_real_but_small(trained_model_and_nominal):
    """The ambiguous-sample calibration guarantees faults in the nominal
    split: without them nominal APFD is NaN (r01 full-run evidence)."""
    _, pred, ny = trained_model_and_nominal
    err = float((pred != ny).mean())
    assert 0.005 < err < 0.30, f"nominal error {err:.3f} outside (0.5%, 30%)"


def test_gini_nominal_apfd_beats_random(trained_model_and_nominal):
    probs, pred, ny = trained_model_and_nominal
    mis = pred != ny
    assert mis.any() and not mis.all()
    gini = ops.softmax_uncertainties(probs)["deep_gini"].numpy()
    order = np.argsort(-gini, kind="stable")
    apfd_gini = apfd_from_order(mis, order)

    rng = np.random.RandomState(0)
    rand_apfds = [
        apfd_from_order(mis, rng.permutation(mis.shape[0])) for _ in range(50)
    ]
    assert apfd_gini > 0.6, f"gini nominal APFD {apfd_gini:.3f} <= 0.6"
    assert apfd_gini > np.mean(rand_apfds) + 0.05, (
        f"gini {apfd_gini:.3f} does not beat random {np.mean(rand_apfds):.3f}"
    )
