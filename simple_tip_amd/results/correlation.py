"""Pairwise statistical comparison of approaches (reference
src/plotters/{eval_apfd_correlation, eval_active_correlation,
correlation_plot}.py).

Pools per-(setting, run) samples across settings, then for every approach
pair computes a two-sided Wilcoxon signed-rank p (Bonferroni-corrected by
C(39, 2)) and a paired Vargha-Delaney A12 folded to [0, 1]
(reference correlation_plot.py:22-45). Emits
results/{exp}_correlation_p.csv / _eff.csv plus the paper's dual-triangle
heatmap figure (p-values upper triangle, effect sizes lower triangle —
reference correlation_plot.py:116-183, Figs. 3-4) as
results/{exp}_correlation.png."""

import logging
import os
from math import comb
from typing import Dict, List

import numpy as np
import pandas as pd
from scipy.stats import wilcoxon

from .. import config
from ..config import ensure_dir
from .active_table import load_runs
from .apfd_table import load_apfd_values
from .common import APPROACHES, CASE_STUDIES

logger = logging.getLogger(__name__)

#: The 9 approaches shown in the paper's correlation figures
#: (reference utils.py:86-99)
CORRELATION_PLOT_APPROACHES = [
    "SNAC_0", "SNAC_0-cam", "NBC_0-cam",
    "dsa", "pc-mdsa", "pc-mlsa",
    "deep_gini", "softmax", "softmax_entropy",
]


def paired_vargha_delaney_a12(x, y) -> float:
    """Paired A12 folded to [0, 1] (reference correlation_plot.py:22-32)."""
    x, y = np.asarray(x), np.asarray(y)
    assert x.shape == y.shape
    same = np.sum(x == y)
    bigger = np.sum(x > y)
    a12 = (bigger + 0.5 * same) / x.size
    return float(2 * abs(a12 - 0.5))


def _pairwise(measurements: Dict[str, Dict[str, float]], approaches: List[str]):
    n = len(approaches)
    p = np.full((n, n), np.nan)
    e = np.full((n, n), np.nan)
    bonferroni = comb(len(APPROACHES), 2)
    for i in range(n - 1):
        for j in range(i + 1, n):
            mi, mj = measurements.get(approaches[i], {}), measurements.get(approaches[j], {})
            keys = sorted(set(mi) & set(mj))
            if not keys:
                continue
            vi = np.array([mi[k] for k in keys])
            vj = np.array([mj[k] for k in keys])
            if np.all(vi == vj):
                continue
            p[i, j] = min(1.0, wilcoxon(vi, vj, alternative="two-sided").pvalue * bonferroni)
            e[i, j] = paired_vargha_delaney_a12(vi, vj)
    return p, e


def _write(exp: str, approaches, p, e):
    ensure_dir("results")
    pd.DataFrame(p, index=approaches, columns=approaches).to_csv(
        os.path.join(config.OUTPUT_FOLDER, "results", f"{exp}_correlation_p.csv")
    )
    pd.DataFrame(e, index=approaches, columns=approaches).to_csv(
        os.path.join(config.OUTPUT_FOLDER, "results", f"{exp}_correlation_eff.csv")
    )
    try:
        import matplotlib  # noqa: F401

        _plot_heatmap(exp, approaches, p, e)
    except ImportError:
        logger.info("matplotlib not available; skipping heatmap figure")


def _plot_heatmap(exp, approaches, p, e):
    """Dual-triangle heatmap: Bonferroni p upper / folded A12 effect lower
    (reference correlation_plot.py:116-183). Pure matplotlib; the reference
    uses two masked seaborn heatmaps on one axes."""
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    n = len(approaches)
    # upper triangle: p (i<j as computed); lower: effect mirrored to (j,i)
    p_mat = np.full((n, n), np.nan)
    e_mat = np.full((n, n), np.nan)
    iu = np.triu_indices(n, 1)
    p_mat[iu] = p[iu]
    e_mat[(iu[1], iu[0])] = e[iu]

    fig, ax = plt.subplots(figsize=(1.1 * n + 2, 1.0 * n + 1.5))
    im_p = ax.imshow(
        np.ma.masked_invalid(p_mat), cmap="Reds_r", vmin=0.0, vmax=1.0
    )
    im_e = ax.imshow(
        np.ma.masked_invalid(e_mat), cmap="Blues", vmin=0.0, vmax=1.0
    )
    ax.set_xticks(range(n))
    ax.set_xticklabels(approaches, rotation=90)
    ax.set_yticks(range(n))
    ax.set_yticklabels(approaches)
    for i in range(n):  # grey diagonal
        ax.add_patch(
            plt.Rectangle((i - 0.5, i - 0.5), 1, 1, color="0.85", lw=0)
        )
    for i in range(n):
        for j in range(n):
            v = p_mat[i, j] if j > i else e_mat[i, j]
            if np.isfinite(v):
                ax.text(
                    j, i, f"{v:.2f}", ha="center", va="center", fontsize=8,
                    color="black",
                )
    cb_p = fig.colorbar(im_p, ax=ax, fraction=0.045, pad=0.02)
    cb_p.set_label("Wilcoxon p (Bonferroni), upper")
    cb_e = fig.colorbar(im_e, ax=ax, fraction=0.045, pad=0.08)
    cb_e.set_label("Vargha-Delaney effect (folded), lower")
    ax.set_title(f"{exp}: pairwise significance and effect size")
    fig.tight_layout()
    fig.savefig(
        os.path.join(config.OUTPUT_FOLDER, "results", f"{exp}_correlation.png"),
        dpi=120,
    )
    plt.close(fig)


def run_apfd(case_studies=None, approaches=None):
    """Pool APFDs over the 8 (cs x ds) settings; write stats CSVs."""
    case_studies = case_studies or CASE_STUDIES
    approaches = approaches or CORRELATION_PLOT_APPROACHES
    measurements: Dict[str, Dict[str, float]] = {a: {} for a in approaches}
    for cs in case_studies:
        for ds in ("nominal", "ood"):
            apfds = load_apfd_values(cs, ds)
            for a in approaches:
                for mid, v in apfds.get(a, {}).items():
                    measurements[a][f"{cs}:{ds}:{mid}"] = v
    p, e = _pairwise(measurements, approaches)
    _write("apfd", approaches, p, e)
    return p, e


def run_active(case_studies=None, approaches=None):
    """Pool active-learning future-split accuracies; write stats CSVs."""
    case_studies = case_studies or CASE_STUDIES
    approaches = approaches or CORRELATION_PLOT_APPROACHES
    measurements: Dict[str, Dict[str, float]] = {a: {} for a in approaches}
    for cs in case_studies:
        runs = load_runs(cs)
        for a in approaches:
            for obs_split, per_model in runs.get(a, {}).items():
                for mid, accs in per_model.items():
                    # the (ds, "future") split is what the paper reports
                    key = (obs_split, "future")
                    if key in accs:
                        measurements[a][f"{cs}:{obs_split}:{mid}"] = accs[key]
    p, e = _pairwise(measurements, approaches)
    _write("active", approaches, p, e)
    return p, e
