import numpy as np
import torch

from simple_tip_amd.engine.model_handler import BaseModel
from simple_tip_amd.models import MnistCNN


def _data(n=40):
    return np.random.RandomState(0).rand(n, 1, 28, 28).astype(np.float32)


def test_get_activations_layers_and_softmax():
    model = MnistCNN().eval()
    bm = BaseModel(model, activation_layers=[1, 3], include_last_layer=True, predict_batch=16)
    x = _data(40)
    outs = bm.get_activations(x)
    assert len(outs) == 3
    assert outs[0].shape == (40, 32, 13, 13)
    assert outs[1].shape == (40, 64, 5, 5)
    probs = outs[2]
    assert probs.shape == (40, 10)
    assert torch.allclose(probs.sum(dim=1), torch.ones(40), atol=1e-5)


def test_argmax_consistent_with_pred():
    model = MnistCNN().eval()
    x = _data(24)
    bm = BaseModel(model, activation_layers=[3], include_last_layer=True, predict_batch=8)
    outs = bm.get_activations(x)
    pred_from_acts = outs[-1].argmax(dim=1).numpy()
    bm2 = BaseModel(model, activation_layers=None, predict_batch=8)
    pred, unc, times = bm2.get_pred_and_uncertainty(x)
    assert np.array_equal(pred, pred_from_acts)


def test_uncertainties_and_times_taxonomy():
    model = MnistCNN().eval()
    bm = BaseModel(model, activation_layers=None, predict_batch=32)
    pred, unc, times = bm.get_pred_and_uncertainty(_data(16))
    for name in ("deep_gini", "softmax", "pcs", "softmax_entropy", "VR"):
        assert name in unc, name
        assert unc[name].shape == (16,)
        assert len(times[name]) == 4  # [setup, pred, quant, cam]
    # mnist model has dropout -> VR present and in [0, 1]
    assert (unc["VR"] >= 0).all() and (unc["VR"] <= 1).all()


def test_no_dropout_model_skips_vr():
    from simple_tip_amd.models import Cifar10CNN

    model = Cifar10CNN().eval()
    bm = BaseModel(model, activation_layers=None, predict_batch=32)
    x = np.random.RandomState(1).rand(8, 3, 32, 32).astype(np.float32)
    _, unc, _ = bm.get_pred_and_uncertainty(x)
    assert "VR" not in unc


def test_batching_invariance():
    model = MnistCNN().eval()
    x = _data(30)
    bm1 = BaseModel(model, activation_layers=None, predict_batch=7)
    bm2 = BaseModel(model, activation_layers=None, predict_batch=30)
    _, u1, _ = bm1.get_pred_and_uncertainty(x)
    _, u2, _ = bm2.get_pred_and_uncertainty(x)
    for k in ("deep_gini", "softmax", "pcs", "softmax_entropy"):
        np.testing.assert_allclose(u1[k], u2[k], rtol=1e-4, atol=1e-6)
