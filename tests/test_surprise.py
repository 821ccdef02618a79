import numpy as np
import pytest
import torch

from simple_tip_amd.core.surprise import (
    DSA,
    LSA,
    MDSA,
    MLSA,
    MultiModalSA,
    SurpriseCoverageMapper,
    _KmeansDiscriminator,
)


def _train_data(n=200, d=8, classes=2, seed=0):
    rng = np.random.RandomState(seed)
    acts = rng.randn(n, d).astype(np.float64)
    preds = rng.randint(0, classes, size=n)
    # separate the classes so per-class structure is meaningful
    acts[preds == 1] += 3.0
    return torch.from_numpy(acts), torch.from_numpy(preds)


@pytest.mark.parametrize(
    "make_sa",
    [
        lambda a, p: MDSA(a),
        lambda a, p: LSA(a, max_features=5),
        lambda a, p: DSA(a, p),
        lambda a, p: MLSA(a, num_components=2),
    ],
    ids=["mdsa", "lsa", "dsa", "mlsa"],
)
def test_ood_scores_higher_and_deterministic(make_sa):
    acts, preds = _train_data()
    sa = make_sa(acts, preds)
    test_id = acts[:50] + 0.01
    test_ood = acts[:50] + 10.0
    s_id = sa(test_id, preds[:50])
    s_ood = sa(test_ood, preds[:50])
    assert s_id.shape == (50,)
    # metamorphic: far-OOD must be more surprising than in-distribution
    assert bool((s_ood > s_id).all())
    # determinism across repeated calls
    s_id2 = sa(test_id, preds[:50])
    assert torch.equal(torch.as_tensor(s_id), torch.as_tensor(s_id2))


def test_duplicated_batch_consistency():
    acts, preds = _train_data()
    sa = DSA(acts, preds)
    single = sa(acts[:20], preds[:20])
    doubled = sa(torch.cat([acts[:20], acts[:20]]), torch.cat([preds[:20], preds[:20]]))
    assert torch.allclose(doubled[:20], single)
    assert torch.allclose(doubled[20:], single)


def test_dsa_positive():
    acts, preds = _train_data()
    sa = DSA(acts, preds)
    vals = sa(acts[:30] + 0.5, preds[:30])
    assert bool((vals > 0).all())


def test_dsa_two_hop_semantics():
    # 1-D fixture where the two-hop d_b is hand-computable
    train = torch.tensor([[0.0], [1.0], [10.0], [11.0]])
    preds = torch.tensor([0, 0, 1, 1])
    sa = DSA(train, preds)
    test = torch.tensor([[2.0]])
    # nearest same-class (label 0) AT is 1.0 -> d_a = 1
    # nearest other-class AT *from 1.0* is 10.0 -> d_b = 9
    val = sa(test, torch.tensor([0]))
    assert val.item() == pytest.approx(1.0 / 9.0)


def test_dsa_subsampling():
    acts, preds = _train_data(n=500)
    sa = DSA(acts, preds, subsampling=0.3)
    assert sa.train_activations.shape[0] == 150


def test_mdsa_matches_numpy_mahalanobis():
    acts, _ = _train_data(n=300, d=6)
    sa = MDSA(acts)
    x = acts[:20] + 1.0
    got = sa(x).numpy()
    mu = acts.numpy().mean(axis=0)
    cov = np.cov(acts.numpy().T, bias=True)
    want = np.einsum(
        "ij,jk,ik->i", x.numpy() - mu, np.linalg.inv(cov), x.numpy() - mu
    )
    np.testing.assert_allclose(got, want, rtol=1e-6)


def test_multimodal_by_class_routes_like_subsas():
    acts, preds = _train_data()
    mm = MultiModalSA.build_by_class(acts, preds, lambda a, p: MDSA(a))
    sub0 = MDSA(acts[preds == 0])
    sub1 = MDSA(acts[preds == 1])
    test = acts[:40]
    tp = preds[:40]
    got = mm(test, tp)
    want = torch.empty(40, dtype=torch.float64)
    want[tp == 0] = sub0(test[tp == 0]).double()
    want[tp == 1] = sub1(test[tp == 1]).double()
    assert torch.allclose(got, want)


def test_kmeans_discriminator_recovers_two_clusters():
    rng = np.random.RandomState(5)
    a = rng.randn(100, 4)
    b = rng.randn(100, 4) + 8.0
    data = torch.from_numpy(np.concatenate([a, b]).astype(np.float32))
    disc = _KmeansDiscriminator(data, potential_k=range(2, 5))
    assert disc.best_k == 2
    labels = disc(data)
    # the two halves get internally-consistent labels
    assert len(torch.unique(labels[:100])) == 1
    assert len(torch.unique(labels[100:])) == 1
    assert labels[0] != labels[150]


def test_multimodal_kmeans_runs():
    acts, preds = _train_data(n=300)
    mm = MultiModalSA.build_with_kmeans(
        acts, preds, lambda a, p: MDSA(a), potential_k=range(2, 4), subsampling=0.5
    )
    vals = mm(acts[:30], preds[:30])
    assert vals.shape == (30,)
    assert torch.isfinite(vals).all()


def test_surprise_coverage_mapper_exact_bits():
    m = SurpriseCoverageMapper(sections=4, upper_bound=8.0)
    vals = torch.tensor([0.0, 1.9, 2.0, 7.999, 8.0, -1.0, 100.0])
    prof = m.get_coverage_profile(vals).to_bool()
    assert prof[0].tolist() == [True, False, False, False]
    assert prof[1].tolist() == [True, False, False, False]
    assert prof[2].tolist() == [False, True, False, False]
    assert prof[3].tolist() == [False, False, False, True]
    # value == upper bound sets no bit (reference half-open intervals)
    assert prof[4].sum() == 0
    assert prof[5].sum() == 0
    assert prof[6].sum() == 0


def test_surprise_coverage_overflow_bucket():
    m = SurpriseCoverageMapper(sections=3, upper_bound=3.0, overflow_bucket=True)
    vals = torch.tensor([0.5, 2.0, 100.0])
    prof = m.get_coverage_profile(vals).to_bool()
    assert prof[0].tolist() == [True, False, False]
    assert prof[1].tolist() == [False, True, False]
    assert prof[2].tolist() == [False, False, True]


def test_lsa_removes_low_variance_features():
    rng = np.random.RandomState(9)
    acts = rng.randn(100, 10)
    acts[:, 3] *= 1e-8  # nearly-constant feature must be dropped
    sa = LSA(torch.from_numpy(acts), max_features=5)
    assert 3 in sa.removed_neurons
    assert len(sa.removed_neurons) >= 5


def test_strict_modes_flag_raises_like_reference():
    """With strict mode on, inputs routed to a mode unseen at fit time RAISE
    (the reference's behaviour, surprise.py:308-315) instead of scoring +inf
    with a warning."""
    import pytest

    from simple_tip_amd.core import surprise as S

    train = torch.randn(40, 6)
    pred = torch.zeros(40, dtype=torch.long)
    pred[20:] = 1  # classes {0, 1} at fit time
    mm = S.MultiModalSA.build_by_class(train, pred, lambda a, p: S.MDSA(a))
    test = torch.randn(8, 6)
    tpred = torch.tensor([0, 1, 2, 0, 1, 2, 2, 0])  # class 2 unseen

    with pytest.warns(UserWarning):
        vals = mm(test, tpred)
    assert torch.isinf(vals[tpred == 2]).all()

    prev = S.set_strict_modes(True)
    try:
        with pytest.raises(ValueError):
            mm(test, tpred)
        dsa = S.DSA(train, pred)
        with pytest.raises(ValueError):
            dsa(test, tpred)
    finally:
        S.set_strict_modes(prev)
