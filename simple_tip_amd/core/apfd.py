"""APFD — Average Percentage of Fault Detection.

Semantics match reference src/core/apfd.py:8-19:
``apfd = 1 - sum(order_of_fault_i) / (k * n) + 1 / (2n)`` where orders are
1-based positions of the faulty inputs in the prioritized order.
"""

from typing import List, Union

import numpy as np


def apfd_from_order(is_fault, index_order: Union[List[int], np.ndarray]) -> float:
    """Compute APFD given a boolean fault vector and a prioritized index order.

    ``index_order[r]`` is the index of the input ranked at position ``r``.
    """
    is_fault = np.asarray(is_fault)
    assert is_fault.ndim == 1, "only unique faults are supported"
    index_order = np.asarray(index_order)
    ordered_faults = is_fault[index_order]
    fault_positions = np.where(ordered_faults != 0)[0]
    k = int(np.count_nonzero(is_fault))
    n = int(is_fault.shape[0])
    if k == 0 or n == 0:
        return float("nan")
    sum_of_fault_orders = np.sum(fault_positions + 1)
    return float(1.0 - (sum_of_fault_orders / (k * n)) + (1.0 / (2 * n)))
