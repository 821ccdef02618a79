"""Results-layer contracts on fabricated artifacts: APFD table values and
the reference's time-column semantics (setup + 2*(pred+quant) + 2*cam)."""

import os
import pickle

import numpy as np
import pytest


@pytest.fixture()
def assets(tmp_path, monkeypatch):
    import simple_tip_amd.config as config

    monkeypatch.setattr(config, "OUTPUT_FOLDER", str(tmp_path))
    (tmp_path / "priorities").mkdir()
    (tmp_path / "times").mkdir()
    (tmp_path / "active_learning").mkdir()
    return tmp_path


def _put(assets, name, arr):
    np.save(assets / "priorities" / name, np.asarray(arr))


def test_apfd_table_values_and_averaging(assets):
    from simple_tip_amd.results import apfd_table

    # two models; 4 inputs; model 0: faults at idx 0,1; model 1: fault at 3
    _put(assets, "mnist_nominal_0_is_misclassified.npy", [1, 1, 0, 0])
    _put(assets, "mnist_nominal_1_is_misclassified.npy", [0, 0, 0, 1])
    # deep_gini: model 0 ranks faults first (apfd .75); model 1 ranks the
    # fault last (apfd = 1 - 4/4 + 1/8 = .125)
    _put(assets, "mnist_nominal_0_uncertainty_deep_gini.npy", [4.0, 3.0, 2.0, 1.0])
    _put(assets, "mnist_nominal_1_uncertainty_deep_gini.npy", [4.0, 3.0, 2.0, 1.0])
    # a cam order artifact
    _put(assets, "mnist_nominal_0_NAC_0_cam_order.npy", [0, 1, 2, 3])

    df = apfd_table.build_dataframe(case_studies=["mnist"])
    gini = df.loc[("uncertainty", "deep_gini"), ("mnist", "nominal")]
    assert float(gini) == pytest.approx((0.75 + 0.125) / 2)
    nac = df.loc[("neuron coverage", "NAC_0-cam"), ("mnist", "nominal")]
    assert float(nac) == pytest.approx(0.75)
    assert df.loc[("surprise", "dsa"), ("mnist", "nominal")] == "n.a."


def test_time_column_semantics(assets):
    """total = mean(setup) + 2*mean(pred) + 2*mean(quant); cam adds 2*cam
    (reference eval_apfd_table.py:176-232)."""
    from simple_tip_amd.results import apfd_table

    _put(assets, "mnist_nominal_0_is_misclassified.npy", [1, 0])
    _put(assets, "mnist_nominal_0_dsa_scores.npy", [1.0, 0.0])
    for ds, times in [("nominal", [10.0, 2.0, 3.0, 1.0]), ("ood", [10.0, 4.0, 5.0, 3.0])]:
        with open(assets / "times" / f"mnist_{ds}_0_dsa", "wb") as f:
            pickle.dump(times, f)
    df = apfd_table.build_dataframe(case_studies=["mnist"])
    # means over the two (ds) entries: setup 10, pred 3, quant 4, cam 2
    assert df.loc[("surprise", "dsa"), ("mnist", "time")] == "24.0s"
    assert df.loc[("surprise", "dsa-cam"), ("mnist", "time")] == "28.0s"


def test_times_ignore_models_beyond_first_10(assets):
    from simple_tip_amd.results import apfd_table

    with open(assets / "times" / "mnist_nominal_3_dsa", "wb") as f:
        pickle.dump([1.0, 1.0, 1.0, 1.0], f)
    with open(assets / "times" / "mnist_nominal_55_dsa", "wb") as f:
        pickle.dump([1000.0, 1000.0, 1000.0, 1000.0], f)
    times = apfd_table._load_times(["mnist"])
    assert ("mnist", "nominal", 3, "dsa") in times
    assert all(k[2] < 10 for k in times)


def test_active_table_delta_vs_random(assets):
    from simple_tip_amd.results import active_table

    def put(mid, metric, split, accs):
        with open(
            assets / "active_learning" / f"mnist_{mid}_{metric}_{split}.pickle", "wb"
        ) as f:
            pickle.dump(accs, f)

    accs_r = {("nominal", "future"): 0.5, ("nominal", "observed"): 0.5,
              ("ood", "future"): 0.4, ("ood", "observed"): 0.4}
    accs_g = {("nominal", "future"): 0.65, ("nominal", "observed"): 0.6,
              ("ood", "future"): 0.5, ("ood", "observed"): 0.45}
    for mid in (0, 1):
        put(mid, "random", "nominal", accs_r)
        put(mid, "deep_gini", "nominal", accs_g)
    df = active_table.build_dataframe(case_studies=["mnist"])
    v = df.loc[("uncertainty", "deep_gini"), ("mnist", "nominal", "nominal-future")]
    assert float(v) == pytest.approx(0.15)
    v = df.loc[("uncertainty", "deep_gini"), ("mnist", "nominal", "ood-observed")]
    assert float(v) == pytest.approx(0.05)


def test_correlation_outputs(assets):
    from simple_tip_amd.results import correlation

    rng = np.random.RandomState(0)
    for mid in range(12):
        mask = rng.rand(50) < 0.4
        _put(assets, f"mnist_nominal_{mid}_is_misclassified.npy", mask)
        _put(assets, f"mnist_nominal_{mid}_uncertainty_deep_gini.npy",
             mask + rng.rand(50) * 0.5)  # correlated with faults
        _put(assets, f"mnist_nominal_{mid}_uncertainty_softmax.npy", rng.rand(50))
    p, e = correlation.run_apfd(
        case_studies=["mnist"], approaches=["deep_gini", "softmax"]
    )
    assert os.path.exists(assets / "results" / "apfd_correlation_p.csv")
    assert 0 <= p[0, 1] <= 1
    assert 0 <= e[0, 1] <= 1


def test_correlation_heatmap_png(tmp_path, monkeypatch):
    """Fig 3/4 parity: the dual-triangle heatmap renders to PNG from
    synthetic measurement dicts (reference
    correlation_plot.py:116-183)."""
    import numpy as np

    from simple_tip_amd import config
    from simple_tip_amd.results import correlation

    monkeypatch.setattr(config, "OUTPUT_FOLDER", str(tmp_path))
    approaches = ["dsa", "pc-lsa", "deep_gini", "softmax"]
    rng = np.random.RandomState(0)
    measurements = {
        a: {f"s:{k}": float(rng.rand()) for k in range(12)} for a in approaches
    }
    p, e = correlation._pairwise(measurements, approaches)
    assert np.isfinite(p[0, 1]) and np.isfinite(e[0, 1])
    correlation._write("testexp", approaches, p, e)
    png = tmp_path / "results" / "testexp_correlation.png"
    assert png.exists() and png.stat().st_size > 5000
    assert (tmp_path / "results" / "testexp_correlation_p.csv").exists()
