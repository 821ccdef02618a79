"""Shared constants for the results layer: canonical approach lists and
category mapping (reference src/plotters/utils.py:21-157)."""

import os
import re
from typing import Dict, List, Tuple

import numpy as np

from .. import config

NUM_RUNS = 100
N_FIRST_MODELS_FOR_TIMES = 10

UNCERTAINTY_APPROACHES = ["deep_gini", "softmax", "pcs", "softmax_entropy", "VR"]

NC_METRICS = [
    "NAC_0.75", "NAC_0", "NBC_0.5", "NBC_0", "NBC_1",
    "SNAC_0.5", "SNAC_0", "SNAC_1", "TKNC_1", "TKNC_2", "TKNC_3", "KMNC_2",
]

SA_METRICS = ["dsa", "pc-lsa", "pc-mdsa", "pc-mlsa", "pc-mmdsa"]

#: All 39 tested approaches (reference utils.py:21-61)
APPROACHES: List[str] = (
    [m + s for m in NC_METRICS for s in ("-cam", "")]
    + [m + s for m in SA_METRICS for s in ("-cam", "")]
    + UNCERTAINTY_APPROACHES
)

CASE_STUDIES = ["mnist", "fmnist", "cifar10", "imdb"]


def category(approach: str) -> str:
    """Approach family, for the table's category index."""
    base = approach[:-4] if approach.endswith("-cam") else approach
    if base in UNCERTAINTY_APPROACHES:
        return "uncertainty"
    if base in SA_METRICS:
        return "surprise"
    return "neuron coverage"


def parse_priority_filename(
    fname: str, case_study: str, ds_name: str
) -> Tuple[int, str]:
    """Split '{cs}_{ds}_{model_id}_{data_type}.npy' -> (model_id, data_type).

    Prefix-based so case-study names containing underscores work."""
    prefix = f"{case_study}_{ds_name}_"
    assert fname.startswith(prefix) and fname.endswith(".npy")
    rest = fname[len(prefix) : -len(".npy")]
    model_id, data_type = rest.split("_", 1)
    return int(model_id), data_type


def iter_priority_files(case_study: str, ds_name: str):
    """Yield (model_id, data_type, path) for every matching artifact."""
    folder = os.path.join(config.OUTPUT_FOLDER, "priorities")
    if not os.path.isdir(folder):
        return
    prefix = f"{case_study}_{ds_name}_"
    for fname in sorted(os.listdir(folder)):
        if not fname.endswith(".npy") or not fname.startswith(prefix):
            continue
        mid, dtype = parse_priority_filename(fname, case_study, ds_name)
        yield mid, dtype, os.path.join(folder, fname)
