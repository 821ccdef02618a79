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


def test_fused_bf16_matches_fp32_within_rounding():
    """The bf16 grouped pairwise kernels (fp32 accumulate) must agree with
    the fp32 path up to bf16 input rounding: distances within ~1%, argmin
    agreement high (flips only between near-ties), and the DSA/LSA score
    ORDERINGS essentially preserved."""
    import numpy as np

    from simple_tip_amd.core.surprise import DSA, LSA, MultiModalSA
    from simple_tip_amd.engine.serving import FusedPrioritizer

    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    train = torch.randn(6000, 512, device=dev)
    pred = torch.randint(0, 10, (6000,), device=dev)
    test = torch.randn(2048, 512, device=dev)
    tpred = torch.randint(0, 10, (2048,), device=dev)

    dsa = DSA(train, pred, device=dev)
    lsa = MultiModalSA.build_by_class(
        train, pred, lambda a, p: LSA(a, max_features=300, device=dev)
    )
    fp32 = FusedPrioritizer(dsa, lsa, dev)
    b16 = FusedPrioritizer(dsa, lsa, dev, pairwise_dtype=torch.bfloat16)
    assert b16.bf16 and not fp32.bf16

    d32, l32 = fp32(test, tpred)
    d16, l16 = b16(test, tpred)
    rel = ((d16 - d32).abs() / d32.abs().clamp_min(1e-3)).median()
    assert float(rel) < 0.02, f"median DSA rel err {float(rel):.4f}"
    # orderings: Spearman-ish check via rank correlation of the scores
    def rankcorr(a, b):
        ra = a.float().argsort().argsort().float()
        rb = b.float().argsort().argsort().float()
        ra = (ra - ra.mean()) / ra.std()
        rb = (rb - rb.mean()) / rb.std()
        return float((ra * rb).mean())

    assert rankcorr(d16, d32) > 0.99
    fin = torch.isfinite(l32) & torch.isfinite(l16)
    assert rankcorr(l16[fin], l32[fin]) > 0.98
