"""Active-learning results table (reference
src/plotters/eval_active_learning_table.py).

Emits /assets/results/active.csv: per (case_study, observed-split,
eval-split), the mean accuracy delta of each TIP's retrained model vs the
``random``-selection baseline (reference semantics:
_relative_active_learning_gains with baseline='random')."""

import logging
import os
import pickle
from typing import Dict, List, Tuple

import numpy as np
import pandas as pd

from .. import config
from ..config import ensure_dir
from .common import APPROACHES, CASE_STUDIES, category

logger = logging.getLogger(__name__)

RANDOM = "random"


def load_runs(case_study: str) -> Dict[str, Dict[str, Dict[int, dict]]]:
    """{approach: {ood_or_nom: {model_id: {(split, obs/fut): acc}}}}."""
    folder = os.path.join(config.OUTPUT_FOLDER, "active_learning")
    res: Dict[str, Dict[str, Dict[int, dict]]] = {}
    if not os.path.isdir(folder):
        return res
    prefix = f"{case_study}_"
    names = APPROACHES + [RANDOM, "original"]
    for fname in os.listdir(folder):
        if not fname.startswith(prefix) or not fname.endswith(".pickle"):
            continue
        rest = fname[len(prefix) : -len(".pickle")]
        mid_s, rest2 = rest.split("_", 1)
        # rest2 = "{metric}_{ood|nominal|na}"
        metric, ood_or_nom = rest2.rsplit("_", 1)
        if metric not in names:
            continue
        with open(os.path.join(folder, fname), "rb") as f:
            data = pickle.load(f)
        res.setdefault(metric, {}).setdefault(ood_or_nom, {})[int(mid_s)] = data
    return res


def build_dataframe(case_studies=None) -> pd.DataFrame:
    case_studies = case_studies or CASE_STUDIES
    eval_splits = [
        ("nominal", "observed"), ("nominal", "future"),
        ("ood", "observed"), ("ood", "future"),
    ]
    cols = pd.MultiIndex.from_tuples(
        [
            (cs, obs, f"{es[0]}-{es[1]}")
            for cs in case_studies
            for obs in ("nominal", "ood")
            for es in eval_splits
        ],
        names=["case_study", "observed_split", "eval_split"],
    )
    rows = [(category(a), a) for a in APPROACHES]
    row_idx = pd.MultiIndex.from_tuples(rows, names=["category", "approach"])
    df = pd.DataFrame(columns=cols, index=row_idx)

    for cs in case_studies:
        runs = load_runs(cs)
        if RANDOM not in runs:
            continue
        for obs_split in ("nominal", "ood"):
            base_runs = runs[RANDOM].get(obs_split, {})
            if not base_runs:
                continue
            for cat, approach in rows:
                a_runs = runs.get(approach, {}).get(obs_split, {})
                common = sorted(set(a_runs) & set(base_runs))
                if not common:
                    continue
                for es in eval_splits:
                    deltas = [
                        a_runs[m][es] - base_runs[m][es] for m in common
                    ]
                    df.loc[(cat, approach), (cs, obs_split, f"{es[0]}-{es[1]}")] = (
                        float(np.mean(deltas))
                    )
    return df


def run(case_studies=None) -> pd.DataFrame:
    """Generate results/active.csv."""
    df = build_dataframe(case_studies)
    ensure_dir("results")
    df.to_csv(os.path.join(config.OUTPUT_FOLDER, "results", "active.csv"))
    logger.info("wrote %s/results/active.csv", config.OUTPUT_FOLDER)
    return df
