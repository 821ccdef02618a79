import numpy as np
import pytest
import torch

from simple_tip_amd.core.kmeans import kmeans_fit, kmeans_predict, silhouette_score


def _blobs(seed=0, n=120, d=6, k=3, sep=8.0):
    rng = np.random.RandomState(seed)
    centers = rng.randn(k, d) * sep
    x = np.concatenate([centers[i] + rng.randn(n // k, d) for i in range(k)])
    y = np.repeat(np.arange(k), n // k)
    return torch.from_numpy(x.astype(np.float32)), y


def test_kmeans_recovers_blobs():
    x, y = _blobs()
    centers, labels, inertia = kmeans_fit(x, 3, n_init=4, seed=0)
    labels = labels.numpy()
    # each true cluster maps to exactly one predicted label
    for c in range(3):
        assert len(np.unique(labels[y == c])) == 1
    assert len(np.unique([labels[y == c][0] for c in range(3)])) == 3
    assert inertia < 2 * x.shape[0] * x.shape[1]


def test_kmeans_deterministic():
    x, _ = _blobs(seed=1)
    c1, l1, i1 = kmeans_fit(x, 3, n_init=3, seed=5)
    c2, l2, i2 = kmeans_fit(x, 3, n_init=3, seed=5)
    assert torch.equal(c1, c2) and torch.equal(l1, l2) and i1 == i2


def test_kmeans_predict_matches_fit_labels():
    x, _ = _blobs(seed=2)
    centers, labels, _ = kmeans_fit(x, 3, n_init=2, seed=0)
    assert torch.equal(kmeans_predict(x, centers), labels)


def test_silhouette_separated_vs_random():
    x, _ = _blobs(seed=3, sep=10.0)
    _, labels, _ = kmeans_fit(x, 3, n_init=2, seed=0)
    good = silhouette_score(x, labels)
    rng = np.random.RandomState(0)
    bad = silhouette_score(x, torch.from_numpy(rng.randint(0, 3, x.shape[0])))
    assert good > 0.7 > bad


def test_silhouette_matches_sklearn():
    from sklearn.metrics import silhouette_score as sk_sil

    x, _ = _blobs(seed=4, sep=3.0)
    _, labels, _ = kmeans_fit(x, 3, n_init=2, seed=0)
    ours = silhouette_score(x, labels)
    ref = sk_sil(x.numpy(), labels.numpy())
    assert ours == pytest.approx(ref, abs=1e-4)


def test_ensemble_spawn_pool():
    from simple_tip_amd.engine.ensemble import run_tasks

    out = run_tasks(_square, [1, 2, 3, 4], num_processes=2)
    assert out == [1, 4, 9, 16]


def _square(x):
    return x * x


def test_pack_conv_fragment_math():
    """_pack_conv reproduces the documented B-fragment gather exactly."""
    from simple_tip_amd.models.resnet_fused import _pack_conv

    torch.manual_seed(0)
    cout, cin = 32, 16
    w = torch.randn(cout, cin, 3, 3)
    pack = _pack_conv(w, 9).float()  # [2, 5, 64, 8]
    k_total = 9 * cin
    w2 = w.permute(2, 3, 1, 0).reshape(k_total, cout)
    for ct in (0, 1):
        for ks in (0, 4):
            for lane in (0, 17, 63):
                g, j = lane >> 4, lane & 15
                for e in (0, 7):
                    k = ks * 32 + g * 8 + e
                    want = w2[k, ct * 16 + j] if k < k_total else 0.0
                    got = pack[ct, ks, lane, e]
                    assert got == pytest.approx(
                        float(torch.tensor(want).to(torch.bfloat16)), abs=1e-6
                    ), (ct, ks, lane, e)


def test_silhouette_ignores_empty_clusters():
    """Empty clusters (kmeans can retain unclaimed centers) must not feed
    b=0 into the min: score must equal sklearn's on the non-empty clusters
    (ADVICE r01)."""
    from sklearn.metrics import silhouette_score as sk_sil

    from simple_tip_amd.core.kmeans import silhouette_score

    torch.manual_seed(3)
    x = torch.randn(40, 5) + torch.cat(
        [torch.zeros(20, 5), torch.full((20, 5), 4.0)]
    )
    labels = torch.cat([torch.zeros(20), torch.full((20,), 2)]).long()
    got = silhouette_score(x, labels)  # cluster id 1 is empty
    want = float(sk_sil(x.numpy(), labels.numpy()))
    assert abs(got - want) < 1e-5
