"""HIP kernel numerics vs the torch/numpy oracle (ops/fallback.py).

All tests are @gpu: they run on a real MI355X via gpurun and at round end.
Random (asymmetric) operands are used throughout so operand/output
transposes cannot slip through (guide §5.4 rule 16)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from simple_tip_amd.ops import fallback


@pytest.fixture(scope="module")
def ext():
    from simple_tip_amd.ops import hip_ops

    return hip_ops


def _rand(m, k, seed=0):
    return torch.from_numpy(np.random.RandomState(seed).randn(m, k).astype(np.float32))


@pytest.mark.parametrize("m,n,k", [(64, 64, 32), (377, 291, 513), (1000, 777, 1600), (130, 1300, 20)])
def test_pairwise_sqdist_matches_fp64(ext, m, n, k):
    a, b = _rand(m, k, 1), _rand(n, k, 2)
    got = ext.pairwise_sqdist(a.cuda(), b.cuda()).cpu().double()
    want = torch.cdist(a.double(), b.double()) ** 2
    scale = want.clamp_min(1.0)
    assert ((got - want).abs() / scale).max() < 1e-4


@pytest.mark.parametrize("m,n,k", [(256, 512, 128), (513, 2050, 300), (70, 15000, 2304)])
def test_rowmin_l2(ext, m, n, k):
    a, b = _rand(m, k, 3), _rand(n, k, 4)
    dist, idx = ext.rowmin_l2(a.cuda(), b.cuda())
    dist, idx = dist.cpu().double(), idx.cpu()
    d64 = torch.cdist(a.double(), b.double())
    want_dist = d64.min(dim=1).values
    assert torch.allclose(dist, want_dist, rtol=1e-4, atol=1e-4)
    # the reported argmin's distance must equal the row minimum
    picked = d64[torch.arange(m), idx]
    assert torch.allclose(picked, want_dist, rtol=1e-4, atol=1e-4)


def test_rowmin_tie_prefers_lowest_index(ext):
    b = torch.zeros(300, 64)
    b[250:] = 5.0  # rows 0..249 identical (all zeros)
    a = torch.zeros(4, 64)
    dist, idx = ext.rowmin_l2(a.cuda(), b.cuda())
    assert (idx.cpu() == 0).all()
    assert torch.allclose(dist.cpu(), torch.zeros(4), atol=1e-5)


@pytest.mark.parametrize("m,n,k", [(128, 256, 64), (413, 5000, 300)])
def test_kde_logsumexp(ext, m, n, k):
    a, b = _rand(m, k, 5) * 0.3, _rand(n, k, 6) * 0.3
    got = ext.kde_logsumexp(a.cuda(), b.cuda()).cpu().double()
    want = fallback.kde_logsumexp(a.double(), b.double())
    assert torch.allclose(got, want, rtol=1e-4, atol=1e-3)


def test_pack_popcount_roundtrip(ext):
    rng = np.random.RandomState(7)
    bools = torch.from_numpy(rng.rand(333, 1337) < 0.3)
    words = ext.pack_bits(bools.cuda())
    want_words = fallback.pack_bits(bools)
    assert torch.equal(words.cpu(), want_words)
    counts = ext.popcount_rows(words)
    assert torch.equal(counts.cpu(), bools.sum(dim=1).long())


def test_nac_profile(ext):
    acts = _rand(200, 500, 8)
    got = ext.nac_profile(acts.cuda(), 0.25)
    want = fallback.nac_profile(acts, 0.25)
    assert torch.equal(got.cpu(), want)


def test_snac_nbc_profiles(ext):
    acts = _rand(150, 300, 9)
    lo = acts.min(dim=0).values - 0.1
    hi = acts.max(dim=0).values - 0.5  # some exceedances
    got = ext.snac_profile(acts.cuda(), hi.cuda())
    want = fallback.snac_profile(acts, hi)
    assert torch.equal(got.cpu(), want)
    got = ext.nbc_profile(acts.cuda(), lo.cuda(), hi.cuda())
    want = fallback.nbc_profile(acts, lo, hi)
    assert torch.equal(got.cpu(), want)


@pytest.mark.parametrize("sections", [2, 5])
def test_kmnc_profile(ext, sections):
    acts = _rand(100, 257, 10)
    mins = acts.min(dim=0).values * 0.8
    maxs = acts.max(dim=0).values * 0.8
    got = ext.kmnc_profile(acts.cuda(), mins.cuda(), maxs.cuda(), sections)
    want = fallback.kmnc_profile(acts, mins, maxs, sections)
    assert torch.equal(got.cpu(), want)


@pytest.mark.parametrize("k", [1, 2, 3])
def test_tknc_profile(ext, k):
    layers = [_rand(80, 1000, 11), _rand(80, 77, 12)]
    got = ext.tknc_profile([l.cuda() for l in layers], k)
    want = fallback.tknc_profile(layers, k)
    assert torch.equal(got.cpu(), want)


def test_bucketize_profile(ext):
    vals = torch.from_numpy(
        np.random.RandomState(13).rand(500).astype(np.float64) * 10
    )
    thr = torch.linspace(0.0, 8.0, 101, dtype=torch.float64)
    got = ext.bucketize_profile(vals.cuda(), thr)
    want = fallback.bucketize_profile(vals, thr)
    assert torch.equal(got.cpu(), want)


def test_cam_order_matches_fallback(ext):
    rng = np.random.RandomState(14)
    profiles = torch.from_numpy(rng.rand(400, 3000) < 0.01)
    scores = torch.from_numpy(rng.rand(400).astype(np.float32))
    words = fallback.pack_bits(profiles)
    want = fallback.cam_order(scores, words, 3000)
    got = ext.cam_order(scores.cuda(), words.cuda(), 3000)
    assert torch.equal(got.cpu(), want)


def test_softmax_scores(ext):
    logits = _rand(1000, 10, 15)
    probs = torch.softmax(logits, dim=1)
    got = ext.softmax_uncertainties(probs.cuda())
    want = fallback.softmax_uncertainties(probs)
    for k in want:
        assert torch.allclose(got[k].cpu(), want[k], rtol=1e-5, atol=1e-6), k


def test_dsa_gpu_matches_cpu():
    from simple_tip_amd.core.surprise import DSA

    rng = np.random.RandomState(16)
    acts = torch.from_numpy(rng.randn(500, 96).astype(np.float32))
    preds = torch.from_numpy(rng.randint(0, 5, 500))
    test = torch.from_numpy(rng.randn(100, 96).astype(np.float32))
    tp = torch.from_numpy(rng.randint(0, 5, 100))
    cpu = DSA(acts, preds)(test, tp)
    gpu = DSA(acts, preds, device="cuda:0")(test.cuda(), tp.cuda())
    assert torch.allclose(gpu.cpu(), cpu.float(), rtol=1e-3, atol=1e-4)


def test_lsa_gpu_matches_cpu():
    from simple_tip_amd.core.surprise import LSA

    rng = np.random.RandomState(17)
    acts = torch.from_numpy(rng.randn(800, 40).astype(np.float64))
    test = torch.from_numpy(rng.randn(120, 40).astype(np.float64))
    cpu = LSA(acts, max_features=30)(test)
    gpu = LSA(acts, max_features=30, device="cuda:0")(test)
    assert torch.allclose(gpu.cpu().double(), cpu, rtol=1e-3, atol=1e-3)


def test_mfma_pairwise_identity_probe():
    """A=I-style probe with ASYMMETRIC B (guide: transpose detection)."""
    from simple_tip_amd.ops import hip_ops

    k = 64
    a = torch.eye(k, dtype=torch.float32)  # rows = unit vectors
    b = torch.zeros(3, k)
    b[0, 5] = 2.0
    b[1, 10] = -1.0
    b[2, 63] = 3.0
    d = hip_ops.pairwise_sqdist(a.cuda(), b.cuda()).cpu()
    want = torch.cdist(a, b) ** 2
    assert torch.allclose(d, want, atol=1e-4)
