import numpy as np
import pytest

from simple_tip_amd.core.apfd import apfd_from_order


def test_all_faults_first():
    is_fault = np.array([1, 1, 0, 0])
    order = [0, 1, 2, 3]
    # faults at positions 1,2 -> 1 - 3/(2*4) + 1/8 = 0.75
    assert apfd_from_order(is_fault, order) == pytest.approx(0.75)


def test_all_faults_last():
    is_fault = np.array([1, 1, 0, 0])
    order = [2, 3, 0, 1]
    # faults at positions 3,4 -> 1 - 7/8 + 1/8 = 0.25
    assert apfd_from_order(is_fault, order) == pytest.approx(0.25)


def test_single_fault_middle():
    is_fault = np.array([0, 1, 0, 0, 0])
    order = [0, 1, 2, 3, 4]
    # fault at position 2 -> 1 - 2/5 + 1/10 = 0.7
    assert apfd_from_order(is_fault, order) == pytest.approx(0.7)


def test_reversal_symmetry():
    rng = np.random.RandomState(0)
    is_fault = rng.rand(100) < 0.3
    order = rng.permutation(100)
    a = apfd_from_order(is_fault, order)
    b = apfd_from_order(is_fault, order[::-1])
    # positions sum to n+1 per fault, so apfd(order) + apfd(reversed) = 1
    assert a + b == pytest.approx(1.0)


def test_no_faults_nan():
    assert np.isnan(apfd_from_order(np.zeros(4), [0, 1, 2, 3]))
