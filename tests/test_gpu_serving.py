"""FusedPrioritizer (grouped segmented kernels) vs the per-class path."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _fit(n=3000, d=256, classes=7, seed=0):
    from simple_tip_amd.core.surprise import DSA, LSA, MultiModalSA

    rng = np.random.RandomState(seed)
    ats = torch.from_numpy(rng.randn(n, d).astype(np.float32))
    pred = torch.from_numpy(rng.randint(0, classes, n))
    dev = torch.device("cuda:0")
    dsa = DSA(ats, pred, device=dev)
    dsa_cpu = DSA(ats, pred)  # independent CPU reference (per-class loop)
    lsa = MultiModalSA.build_by_class(
        ats, pred, lambda a, p: LSA(a, max_features=64, device=dev)
    )
    return dsa, dsa_cpu, lsa, dev


def test_fused_matches_per_class():
    from simple_tip_amd.engine.serving import FusedPrioritizer

    dsa, dsa_cpu, lsa, dev = _fit()
    fused = FusedPrioritizer(dsa, lsa, dev)
    assert fused.lsa_ready

    rng = np.random.RandomState(1)
    test = torch.from_numpy(rng.randn(777, 256).astype(np.float32)).to(dev)
    tp = torch.from_numpy(rng.randint(0, 7, 777)).to(dev)

    d_fused, l_fused = fused(test, tp)
    d_ref = dsa_cpu(test.cpu(), tp.cpu())
    l_ref = lsa(test, tp)

    assert torch.allclose(d_fused.cpu(), d_ref.float(), rtol=1e-3, atol=1e-4)
    assert torch.allclose(
        l_fused.cpu().double(), l_ref.cpu().double(), rtol=1e-3, atol=1e-3
    )


def test_fused_skewed_classes():
    from simple_tip_amd.engine.serving import FusedPrioritizer

    dsa, dsa_cpu, lsa, dev = _fit(seed=2)
    fused = FusedPrioritizer(dsa, lsa, dev)
    rng = np.random.RandomState(3)
    test = torch.from_numpy(rng.randn(500, 256).astype(np.float32)).to(dev)
    # heavy skew incl. an absent class
    tp = torch.from_numpy(
        np.concatenate([np.full(450, 3), rng.randint(0, 2, 50)])
    ).to(dev)
    d_fused, l_fused = fused(test, tp)
    d_ref = dsa_cpu(test.cpu(), tp.cpu())
    assert torch.allclose(d_fused.cpu(), d_ref.float(), rtol=1e-3, atol=1e-4)
    l_ref = lsa(test, tp)
    assert torch.allclose(
        l_fused.cpu().double(), l_ref.cpu().double(), rtol=1e-3, atol=1e-3
    )


def test_fused_degraded_classes():
    """Classes whose KDE can't fit (singleton class) degrade to the
    reference's constant scores, same as the per-class path."""
    from simple_tip_amd.core.surprise import DSA, LSA, MultiModalSA
    from simple_tip_amd.engine.serving import FusedPrioritizer

    rng = np.random.RandomState(7)
    ats = torch.from_numpy(rng.randn(400, 64).astype(np.float32))
    pred = torch.from_numpy(rng.randint(0, 3, 400))
    pred[0] = 3  # class 3 has exactly one training sample
    dev = torch.device("cuda:0")
    dsa = DSA(ats, pred, device=dev)
    lsa = MultiModalSA.build_by_class(
        ats, pred, lambda a, p: LSA(a, max_features=32, device=dev)
    )
    fused = FusedPrioritizer(dsa, lsa, dev)
    assert fused.lsa_ready
    test = torch.from_numpy(rng.randn(100, 64).astype(np.float32)).to(dev)
    tp = torch.from_numpy(rng.randint(0, 4, 100)).to(dev)
    d_f, l_f = fused(test, tp)
    l_ref = lsa(test, tp)
    assert torch.allclose(
        l_f.cpu().double(), l_ref.cpu().double(), rtol=1e-3, atol=1e-3
    )


def test_fused_narrow_features():
    """Grouped kernels with a narrow AT width (IMDB-shaped, D=20, K%4 != 0)."""
    from simple_tip_amd.core.surprise import DSA, LSA, MultiModalSA
    from simple_tip_amd.engine.serving import FusedPrioritizer

    rng = np.random.RandomState(11)
    ats = torch.from_numpy(rng.randn(900, 20).astype(np.float32))
    pred = torch.from_numpy(rng.randint(0, 2, 900))
    dev = torch.device("cuda:0")
    dsa = DSA(ats, pred, device=dev)
    lsa = MultiModalSA.build_by_class(
        ats, pred, lambda a, p: LSA(a, max_features=15, device=dev)
    )
    dsa_cpu = DSA(ats, pred)
    fused = FusedPrioritizer(dsa, lsa, dev)
    test = torch.from_numpy(rng.randn(333, 20).astype(np.float32)).to(dev)
    tp = torch.from_numpy(rng.randint(0, 2, 333)).to(dev)
    d_f, l_f = fused(test, tp)
    d_ref = dsa_cpu(test.cpu(), tp.cpu())
    l_ref = lsa(test, tp)
    assert torch.allclose(d_f.cpu(), d_ref.float(), rtol=1e-3, atol=1e-4)
    assert torch.allclose(
        l_f.cpu().double(), l_ref.cpu().double(), rtol=1e-3, atol=1e-3
    )


def test_fused_determinism():
    from simple_tip_amd.engine.serving import FusedPrioritizer

    dsa, dsa_cpu, lsa, dev = _fit(seed=4)
    fused = FusedPrioritizer(dsa, lsa, dev)
    rng = np.random.RandomState(5)
    test = torch.from_numpy(rng.randn(300, 256).astype(np.float32)).to(dev)
    tp = torch.from_numpy(rng.randint(0, 7, 300)).to(dev)
    d1, l1 = fused(test, tp)
    d2, l2 = fused(test, tp)
    assert torch.equal(d1, d2) and torch.equal(l1, l2)
