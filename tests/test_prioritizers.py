import numpy as np
import torch

from simple_tip_amd.core.prioritizers import cam, ctm


def test_ctm_descending():
    scores = np.array([0.1, 0.9, 0.5, 0.9])
    order = list(ctm(scores))
    # descending, stable on ties (index 1 before 3)
    assert order == [1, 3, 2, 0]


def test_cam_hand_example():
    # rows: coverage profiles over 6 columns
    profiles = np.array(
        [
            [1, 1, 1, 0, 0, 0],  # covers 3
            [0, 0, 0, 1, 1, 0],  # covers 2 new after row 0
            [1, 1, 0, 0, 0, 0],  # subset of row 0
            [0, 0, 0, 0, 0, 1],  # covers the last col
        ],
        dtype=bool,
    )
    scores = np.array([3.0, 2.0, 2.5, 1.0])
    order = list(cam(scores, profiles))
    # greedy: row0 (3 new), row1 (2 new), row3 (1 new); leftover row2 by score
    assert order == [0, 1, 3, 2]


def test_cam_tie_prefers_first_index():
    profiles = np.array([[1, 0], [0, 1]], dtype=bool)
    scores = np.array([0.0, 0.0])
    assert list(cam(scores, profiles))[:1] == [0]


def test_cam_leftovers_by_score():
    profiles = np.zeros((4, 3), dtype=bool)
    profiles[0, 0] = True
    scores = np.array([0.1, 5.0, 1.0, 3.0])
    order = list(cam(scores, profiles))
    assert order == [0, 1, 3, 2]


def _cam_reference(scores, profiles):
    """Brute-force greedy max-cover oracle on bool arrays."""
    profiles = profiles.copy()
    scores = np.asarray(scores, dtype=float)
    n = profiles.shape[0]
    yielded = []
    used = np.zeros(n, dtype=bool)
    while True:
        counts = profiles.sum(axis=1)
        counts[used] = -1  # a used row can never be re-picked
        nxt = int(np.argmax(counts))
        if counts[nxt] <= 0:
            break
        yielded.append(nxt)
        used[nxt] = True
        cols = profiles[nxt].nonzero()[0]
        profiles[:, cols] = False
    left = np.where(~used)[0]
    left = left[np.argsort(-scores[left], kind="stable")]
    return yielded + [int(i) for i in left]


def test_cam_fuzz_matches_bruteforce():
    rng = np.random.RandomState(42)
    for _ in range(20):
        n = rng.randint(2, 60)
        k = rng.randint(1, 200)
        profiles = rng.rand(n, k) < rng.uniform(0.02, 0.4)
        scores = np.round(rng.rand(n), 3)
        got = list(cam(scores, profiles))
        want = _cam_reference(scores, profiles)
        assert got == want
        assert sorted(got) == list(range(n))  # complete and unique


def test_cam_coverage_increments_weakly_decreasing():
    rng = np.random.RandomState(7)
    profiles = rng.rand(100, 500) < 0.05
    scores = rng.rand(100)
    order = list(cam(scores, profiles))
    covered = np.zeros(500, dtype=bool)
    increments = []
    for i in order:
        new = int((profiles[i] & ~covered).sum())
        increments.append(new)
        covered |= profiles[i]
    # greedy phase increments weakly decrease until they hit 0
    greedy = [x for x in increments if x > 0]
    assert all(a >= b for a, b in zip(greedy, greedy[1:]))


def test_cam_accepts_multidim_profiles():
    rng = np.random.RandomState(3)
    profiles = rng.rand(10, 5, 4) < 0.3
    scores = rng.rand(10)
    order = list(cam(scores, profiles))
    assert sorted(order) == list(range(10))
