"""Property-based tests (hypothesis) for the core invariants."""

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from simple_tip_amd.core.apfd import apfd_from_order
from simple_tip_amd.core.bitmap import BitProfile
from simple_tip_amd.core.prioritizers import cam, ctm


@settings(max_examples=40, deadline=None)
@given(
    n=st.integers(2, 40),
    seed=st.integers(0, 10_000),
    p_fault=st.floats(0.1, 0.9),
)
def test_apfd_bounds_and_best_order(n, seed, p_fault):
    rng = np.random.RandomState(seed)
    is_fault = rng.rand(n) < p_fault
    if not is_fault.any():
        is_fault[0] = True
    order = rng.permutation(n)
    v = apfd_from_order(is_fault, order)
    assert 0.0 <= v <= 1.0
    # faults-first ordering maximises APFD over any permutation
    best = np.concatenate([np.where(is_fault)[0], np.where(~is_fault)[0]])
    assert apfd_from_order(is_fault, best) >= v - 1e-12


@settings(max_examples=30, deadline=None)
@given(n=st.integers(1, 60), k=st.integers(1, 300), seed=st.integers(0, 9999))
def test_pack_roundtrip_and_popcount(n, k, seed):
    rng = np.random.RandomState(seed)
    b = torch.from_numpy(rng.rand(n, k) < rng.uniform(0.05, 0.95))
    prof = BitProfile.from_bool(b)
    assert torch.equal(prof.to_bool(), b)
    assert torch.equal(prof.popcount(), b.sum(dim=1).long())


@settings(max_examples=25, deadline=None)
@given(
    n=st.integers(2, 50),
    k=st.integers(1, 400),
    density=st.floats(0.01, 0.3),
    seed=st.integers(0, 9999),
)
def test_cam_properties(n, k, density, seed):
    rng = np.random.RandomState(seed)
    profiles = rng.rand(n, k) < density
    scores = rng.rand(n)
    order = list(cam(scores, profiles))
    # complete permutation
    assert sorted(order) == list(range(n))
    # greedy-phase increments weakly decrease
    covered = np.zeros(k, dtype=bool)
    incs = []
    for i in order:
        incs.append(int((profiles[i] & ~covered).sum()))
        covered |= profiles[i]
    greedy = [x for x in incs if x > 0]
    assert all(a >= b for a, b in zip(greedy, greedy[1:]))
    # the first pick covers the maximum coverable
    assert incs[0] == profiles.sum(axis=1).max()


@settings(max_examples=30, deadline=None)
@given(n=st.integers(1, 100), seed=st.integers(0, 9999))
def test_ctm_is_sorted(n, seed):
    scores = np.random.RandomState(seed).rand(n)
    order = list(ctm(scores))
    vals = scores[order]
    assert all(a >= b for a, b in zip(vals, vals[1:]))
