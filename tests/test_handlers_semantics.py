"""Handler-level contracts: the reference's 12 coverage metrics with
time-debit accounting, and the surprise handler's result taxonomy."""

import numpy as np
import pytest
import torch

from simple_tip_amd.engine.coverage_handler import CoverageWorker
from simple_tip_amd.engine.model_handler import BaseModel
from simple_tip_amd.engine.surprise_handler import SurpriseHandler
from simple_tip_amd.models import MnistCNN


@pytest.fixture(scope="module")
def worker():
    torch.manual_seed(0)
    model = MnistCNN().eval()
    train = np.random.RandomState(0).rand(80, 1, 28, 28).astype(np.float32)
    return CoverageWorker(
        BaseModel(model, [0, 1, 2, 3], device=torch.device("cpu"), predict_batch=32),
        train,
    )


def test_twelve_metrics_configured(worker):
    assert set(worker.metrics) == {
        "NBC_0", "NBC_0.5", "NBC_1", "SNAC_0", "SNAC_0.5", "SNAC_1",
        "NAC_0", "NAC_0.75", "TKNC_1", "TKNC_2", "TKNC_3", "KMNC_2",
    }


def test_time_debits(worker):
    """Reference handler_coverage.py:49-101: metrics that consume the
    aggregate-statistics pass inherit its cost as a setup debit; NAC/TKNC
    need no statistics and carry (almost) no setup."""
    st = worker.setup_times
    assert st["NBC_0"] > st["NAC_0"]
    assert st["KMNC_2"] > st["NAC_0"]
    assert st["SNAC_1"] > st["TKNC_1"]
    # NBC debit includes the welford bucket that KMNC's does not
    assert st["NBC_0"] >= st["KMNC_2"] - 1e-6


def test_evaluate_all_shapes_and_times(worker):
    test = np.random.RandomState(1).rand(48, 1, 28, 28).astype(np.float32)
    times, scores, cam_orders = worker.evaluate_all(test, "nominal")
    for m in worker.metrics:
        assert scores[m].shape == (48,)
        assert sorted(cam_orders[m]) == list(range(48))
        assert len(times[m]) == 4  # [setup, pred, quant, cam]
        assert all(t >= 0 for t in times[m])
    # NAC_0 on relu'd conv outputs: most neurons > 0 for some input
    assert scores["NAC_0"].max() > 0


def test_surprise_handler_taxonomy():
    torch.manual_seed(1)
    model = MnistCNN().eval()
    rng = np.random.RandomState(2)
    train = rng.rand(120, 1, 28, 28).astype(np.float32)
    nominal = rng.rand(40, 1, 28, 28).astype(np.float32)
    sh = SurpriseHandler(model, sa_layers=[3], training_dataset=train,
                         device=torch.device("cpu"), predict_batch=32)
    res = sh.evaluate_all({"nominal": nominal})
    assert set(res) == {"dsa", "pc-lsa", "pc-mdsa", "pc-mlsa", "pc-mmdsa"}
    for name, per_ds in res.items():
        scores, cam_order, times = per_ds["nominal"]
        assert scores.shape == (40,)
        assert sorted(cam_order.tolist()) == list(range(40))
        assert len(times) == 4
        # setup time includes the shared train-AT pass
        assert times[0] >= sh.train_at_timer.get() - 1e-6
