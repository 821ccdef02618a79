import numpy as np
import torch

from simple_tip_amd.core.bitmap import BitProfile


def test_pack_unpack_roundtrip():
    rng = np.random.RandomState(0)
    for k in [1, 7, 63, 64, 65, 130, 1000]:
        b = torch.from_numpy(rng.rand(17, k) < 0.3)
        prof = BitProfile.from_bool(b)
        assert prof.words.shape == (17, (k + 63) // 64)
        assert torch.equal(prof.to_bool(), b)


def test_popcount_matches_sum():
    rng = np.random.RandomState(1)
    b = torch.from_numpy(rng.rand(50, 301) < 0.5)
    prof = BitProfile.from_bool(b)
    assert torch.equal(prof.popcount(), b.sum(dim=1).long())


def test_bit_layout_lsb_first():
    # column j maps to bit j of word j//64 (LSB-first)
    b = torch.zeros(1, 128, dtype=torch.bool)
    b[0, 0] = True
    b[0, 65] = True
    prof = BitProfile.from_bool(b)
    w = prof.words.numpy().view(np.uint64)
    assert w[0, 0] == 1
    assert w[0, 1] == 2


def test_cat():
    b1 = torch.tensor([[True, False, True]])
    b2 = torch.tensor([[False, True, True]])
    p = BitProfile.cat([BitProfile.from_bool(b1), BitProfile.from_bool(b2)])
    assert p.n == 2 and p.nbits == 3
    assert torch.equal(p.to_bool(), torch.cat([b1, b2]))
