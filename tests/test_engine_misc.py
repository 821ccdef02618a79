import os
import pickle

import numpy as np
import pytest
import torch

from simple_tip_amd.engine.aggregate_statistics import (
    AggregateStatisticsCollector,
    WelfordState,
)


def test_welford_matches_numpy():
    rng = np.random.RandomState(0)
    data = rng.randn(500, 7)
    w = WelfordState(7, "cpu")
    for s in range(0, 500, 64):
        w.add_batch(torch.from_numpy(data[s : s + 64]))
    np.testing.assert_allclose(
        w.var_sample().numpy(), data.var(axis=0, ddof=1), rtol=1e-10
    )
    np.testing.assert_allclose(w.mean.numpy(), data.mean(axis=0), rtol=1e-10)


def test_welford_merge():
    rng = np.random.RandomState(1)
    data = rng.randn(300, 4)
    a, b = WelfordState(4, "cpu"), WelfordState(4, "cpu")
    a.add_batch(torch.from_numpy(data[:100]))
    b.add_batch(torch.from_numpy(data[100:]))
    a.merge(b)
    np.testing.assert_allclose(
        a.var_sample().numpy(), data.var(axis=0, ddof=1), rtol=1e-10
    )


def test_aggregate_collector_timers_and_stats():
    rng = np.random.RandomState(2)
    layers = [rng.randn(40, 3, 5).astype(np.float32), rng.randn(40, 8).astype(np.float32)]
    agg = AggregateStatisticsCollector()
    for s in range(0, 40, 16):
        agg.track([torch.from_numpy(l[s : s + 16]) for l in layers])
    mins, maxs, stds = agg.get()
    assert len(mins) == 2
    np.testing.assert_allclose(
        mins[0].numpy(), layers[0].reshape(40, -1).min(axis=0), rtol=1e-6
    )
    np.testing.assert_allclose(
        maxs[1].numpy(), layers[1].max(axis=0), rtol=1e-6
    )
    np.testing.assert_allclose(
        stds[0].numpy(),
        layers[0].reshape(40, -1).std(axis=0, ddof=1),
        rtol=1e-5,
    )
    # timers populated
    assert agg.min_timer.get() >= 0 and agg.welford_timer.get() >= 0
    with pytest.raises(RuntimeError):
        agg.track([torch.from_numpy(l[:4]) for l in layers])


def test_activation_persistor_layout(tmp_path, monkeypatch):
    import simple_tip_amd.config as config
    import simple_tip_amd.engine.activation_persistor as ap

    monkeypatch.setattr(config, "OUTPUT_FOLDER", str(tmp_path))
    from simple_tip_amd.models import MnistCNN

    model = MnistCNN().eval()
    data = np.random.RandomState(0).rand(150, 1, 28, 28).astype(np.float32)
    ap.persist("mnist", 0, model, {"nominal": data}, num_layers=4)
    base = tmp_path / "activations" / "mnist" / "model_0" / "nominal"
    assert (base / "layer_0" / "badge_0.npy").exists()
    assert (base / "layer_3" / "badge_1.npy").exists()
    arr = np.load(base / "layer_3" / "badge_0.npy")
    assert arr.shape == (100, 64, 5, 5)  # BADGE_SIZE=100
    arr2 = np.load(base / "layer_3" / "badge_1.npy")
    assert arr2.shape == (50, 64, 5, 5)


def test_ensemble_save_load(tmp_path, monkeypatch):
    import simple_tip_amd.config as config

    monkeypatch.setattr(config, "OUTPUT_FOLDER", str(tmp_path))
    from simple_tip_amd.engine import ensemble
    from simple_tip_amd.models import MnistCNN

    m = MnistCNN()
    ensemble.save_model("mnist", 3, m)
    m2 = ensemble.load_model("mnist", 3, MnistCNN)
    for p1, p2 in zip(m.parameters(), m2.parameters()):
        assert torch.equal(p1, p2)
    assert not m2.training  # loaded in eval mode


def test_reproduction_cli_eval_phase(tmp_path, monkeypatch):
    monkeypatch.setenv("TIP_ASSETS_DIR", str(tmp_path))
    import simple_tip_amd.config as config

    monkeypatch.setattr(config, "OUTPUT_FOLDER", str(tmp_path))
    from typer.testing import CliRunner
    import typer

    import reproduction

    app = typer.Typer()
    app.command()(reproduction.main)
    runner = CliRunner()
    res = runner.invoke(
        app, ["--phase", "evaluation", "--eval-type", "test_prio", "--yes"]
    )
    assert res.exit_code == 0, res.output
    assert (tmp_path / "results" / "apfds.csv").exists()
