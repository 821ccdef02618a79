"""APFD results table (reference src/plotters/eval_apfd_table.py).

Emits /assets/results/apfds.csv — rows (category, approach), columns
(case_study, {nominal, ood, time}) — and a LaTeX excerpt of the paper
approaches. Time semantics match the reference: mean over the first 10
models of setup + 2*(pred + quant), CAM variants add + 2*cam
(eval_apfd_table.py:176-232)."""

import logging
import os
import pickle
import warnings
from typing import Dict

import numpy as np
import pandas as pd

from .. import config
from ..config import ensure_dir
from ..core.apfd import apfd_from_order
from .common import (
    APPROACHES,
    CASE_STUDIES,
    NUM_RUNS,
    N_FIRST_MODELS_FOR_TIMES,
    category,
    iter_priority_files,
)

logger = logging.getLogger(__name__)
TIME_COL = "time"

PAPER_APPROACHES = [
    "NAC_0.75-cam", "NAC_0.75", "NBC_0-cam", "NBC_0", "SNAC_0-cam", "SNAC_0",
    "TKNC_1-cam", "KMNC_2", "dsa", "pc-lsa", "pc-mdsa", "pc-mlsa", "pc-mmdsa",
    "deep_gini", "softmax", "pcs", "softmax_entropy", "VR",
]


def load_apfd_values(case_study: str, ds_name: str) -> Dict[str, Dict[int, float]]:
    """{approach: {model_id: apfd}} from the priorities artifacts."""
    misclassifications: Dict[int, np.ndarray] = {}
    orders: Dict[tuple, np.ndarray] = {}
    for mid, dtype, path in iter_priority_files(case_study, ds_name):
        if mid >= NUM_RUNS:
            continue
        if dtype == "is_misclassified":
            misclassifications[mid] = np.load(path)
        elif dtype.endswith("_cam_order"):
            metric = dtype[: -len("_cam_order")] + "-cam"
            orders[(metric, mid)] = np.load(path)
        elif dtype.startswith("uncertainty_"):
            metric = dtype[len("uncertainty_") :]
            scores = np.load(path)
            orders[(metric, mid)] = np.argsort(-scores, kind="stable")
        elif dtype.endswith("_scores"):
            metric = dtype[: -len("_scores")]
            scores = np.load(path)
            orders[(metric, mid)] = np.argsort(-scores, kind="stable")

    apfds: Dict[str, Dict[int, float]] = {}
    for (metric, mid), order in orders.items():
        if metric not in APPROACHES:
            continue
        if mid not in misclassifications:
            continue
        apfds.setdefault(metric, {})[mid] = apfd_from_order(
            misclassifications[mid], order
        )
    return apfds


def _load_times(case_studies) -> Dict[tuple, list]:
    """{(cs, ds, model_id, metric): [setup, pred, quant(, cam)]} pickles,
    first 10 models only."""
    folder = os.path.join(config.OUTPUT_FOLDER, "times")
    res = {}
    if not os.path.isdir(folder):
        return res
    for fname in os.listdir(folder):
        for cs in case_studies:
            for ds in ("nominal", "ood"):
                prefix = f"{cs}_{ds}_"
                if not fname.startswith(prefix):
                    continue
                rest = fname[len(prefix) :]
                mid_s, metric = rest.split("_", 1)
                mid = int(mid_s)
                if mid >= N_FIRST_MODELS_FOR_TIMES:
                    continue
                with open(os.path.join(folder, fname), "rb") as f:
                    res[(cs, ds, mid, metric)] = pickle.load(f)
                break
    return res


def _fill_times(df: pd.DataFrame, case_studies) -> pd.DataFrame:
    times = _load_times(case_studies)
    for cs in case_studies:
        metrics = {k[3] for k in times if k[0] == cs}
        for metric in metrics:
            vals = [v for k, v in times.items() if k[0] == cs and k[3] == metric]
            setup = float(np.mean([v[0] for v in vals]))
            pred = float(np.mean([v[1] for v in vals]))
            quant = float(np.mean([v[2] for v in vals]))
            total = setup + 2 * (pred + quant)
            if metric in APPROACHES:
                df.loc[(category(metric), metric), (cs, TIME_COL)] = f"{total:.1f}s"
            cam_metric = f"{metric}-cam"
            if cam_metric in APPROACHES and all(len(v) >= 4 for v in vals):
                cam = float(np.mean([v[3] for v in vals]))
                df.loc[(category(cam_metric), cam_metric), (cs, TIME_COL)] = (
                    f"{total + 2 * cam:.1f}s"
                )
    return df


def build_dataframe(case_studies=None) -> pd.DataFrame:
    case_studies = case_studies or CASE_STUDIES
    col_idx = pd.MultiIndex.from_product([case_studies, ["nominal", "ood", TIME_COL]])
    rows = [(category(a), a) for a in APPROACHES]
    row_idx = pd.MultiIndex.from_tuples(rows, names=["category", "approach"])
    df = pd.DataFrame(columns=col_idx, index=row_idx)
    for cs in case_studies:
        for ds in ("nominal", "ood"):
            apfds = load_apfd_values(cs, ds)
            for cat, approach in rows:
                if approach in apfds and apfds[approach]:
                    df.loc[(cat, approach), (cs, ds)] = float(
                        np.mean(list(apfds[approach].values()))
                    )
                else:
                    df.loc[(cat, approach), (cs, ds)] = "n.a."
    return _fill_times(df, case_studies)


def run(case_studies=None) -> pd.DataFrame:
    """Generate results/apfds.csv (+ LaTeX paper excerpt)."""
    df = build_dataframe(case_studies)
    ensure_dir("results")
    df.to_csv(os.path.join(config.OUTPUT_FOLDER, "results", "apfds.csv"))
    try:
        paper = df.iloc[
            df.index.get_level_values("approach").isin(PAPER_APPROACHES)
        ]
        with open(
            os.path.join(config.OUTPUT_FOLDER, "results", "apfd_paper_table.tex"), "w"
        ) as f:
            f.write(paper.to_latex(multicolumn_format="c", multirow=True))
    except Exception as e:  # noqa: BLE001 - latex formatting is best-effort
        warnings.warn(f"LaTeX table generation failed: {e}")
    logger.info("wrote %s/results/apfds.csv", config.OUTPUT_FOLDER)
    return df
