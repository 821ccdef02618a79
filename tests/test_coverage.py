import numpy as np
import torch

from simple_tip_amd.core.neuron_coverage import KMNC, NAC, NBC, SNAC, TKNC

# Tiny hand-computable fixture: two samples, two "layers" (3 + 2 neurons).
LAYER_A = torch.tensor([[0.0, 0.5, 1.0], [2.0, -1.0, 0.25]])
LAYER_B = torch.tensor([[10.0, -10.0], [0.0, 0.0]])
ACTS = [LAYER_A, LAYER_B]

MINS = [torch.tensor([0.0, 0.0, 0.0]), torch.tensor([-1.0, -1.0])]
MAXS = [torch.tensor([1.0, 1.0, 1.0]), torch.tensor([1.0, 1.0])]
STDS = [torch.tensor([0.5, 0.5, 0.5]), torch.tensor([1.0, 1.0])]


def test_nac():
    scores, prof = NAC(cov_threshold=0.0)(ACTS)
    want = torch.tensor(
        [
            [False, True, True, True, False],
            [True, False, True, False, False],
        ]
    )
    assert torch.equal(prof.to_bool(), want)
    assert scores.tolist() == [3, 2]


def test_nac_threshold():
    scores, prof = NAC(cov_threshold=0.75)(ACTS)
    want = torch.tensor(
        [
            [False, False, True, True, False],
            [True, False, False, False, False],
        ]
    )
    assert torch.equal(prof.to_bool(), want)
    assert scores.tolist() == [2, 1]


def test_snac():
    # bounds = max + 1*std = [1.5,1.5,1.5, 2,2]
    scores, prof = SNAC(maxs=MAXS, stds=STDS, scaler=1.0)(ACTS)
    want = torch.tensor(
        [
            [False, False, False, True, False],
            [True, False, False, False, False],
        ]
    )
    assert torch.equal(prof.to_bool(), want)
    assert scores.tolist() == [1, 1]


def test_nbc():
    # lower = min - 0.5*std = [-0.25,-0.25,-0.25, -1.5,-1.5]
    # upper = max + 0.5*std = [1.25,1.25,1.25, 1.5,1.5]
    scores, prof = NBC(mins=MINS, maxs=MAXS, stds=STDS, scaler=0.5)(ACTS)
    b = prof.to_bool().reshape(2, 5, 2)
    # sample 0: only layer_b neuron 0 (10 >= 1.5) upper, neuron 1 (-10 <= -1.5) lower
    assert b[0].nonzero().tolist() == [[3, 1], [4, 0]]
    # sample 1: layer_a neuron 0 (2 >= 1.25) upper, neuron 1 (-1 <= -0.25) lower
    assert b[1].nonzero().tolist() == [[0, 1], [1, 0]]
    assert scores.tolist() == [2, 2]


def test_kmnc():
    # 2 sections over [min, max]; jumps: layer_a 0.5 each, layer_b 1.0 each
    scores, prof = KMNC(mins=MINS, maxs=MAXS, sections=2)(ACTS)
    b = prof.to_bool().reshape(2, 5, 2)
    # sample 0: a=[0(.s0), .5(s1), 1.0(=max: none)], b=[10(out), -10(out)]
    assert b[0].nonzero().tolist() == [[0, 0], [1, 1]]
    # sample 1: a=[2(out), -1(out), .25(s0)], b=[0(s1), 0(s1)]
    assert b[1].nonzero().tolist() == [[2, 0], [3, 1], [4, 1]]
    assert scores.tolist() == [2, 3]


def test_tknc():
    scores, prof = TKNC(top_neurons=1)(ACTS)
    want = torch.tensor(
        [
            [False, False, True, True, False],
            [True, False, False, True, False],
        ]
    )
    # layer_b sample 1 has a tie (0.0, 0.0) -> accept either winner
    got = prof.to_bool()
    assert torch.equal(got[:, :3], want[:, :3])
    assert got[0, 3:].tolist() == [True, False]
    assert got[1, 3:].sum() == 1
    assert scores.tolist() == [2, 2]


def test_tknc_k2():
    scores, prof = TKNC(top_neurons=2)(ACTS)
    got = prof.to_bool()
    assert got[0, :3].tolist() == [False, True, True]
    assert got[1, :3].tolist() == [True, False, True]
    assert scores.tolist() == [4, 4]


def test_profiles_on_random_data_match_manual():
    rng = np.random.RandomState(0)
    acts = [torch.from_numpy(rng.randn(20, 13).astype(np.float32))]
    mins = [acts[0].min(dim=0).values]
    maxs = [acts[0].max(dim=0).values]
    stds = [acts[0].std(dim=0)]
    s, p = NBC(mins, maxs, stds, scaler=0.0)(acts)
    manual = torch.stack(
        [acts[0] <= mins[0], acts[0] >= maxs[0]], dim=2
    ).reshape(20, -1)
    assert torch.equal(p.to_bool(), manual)
    assert torch.equal(s, manual.sum(dim=1).long())
