"""Active-learning experiment: select-by-TIP, retrain, re-evaluate.

Capability parity with reference src/dnn_test_prio/eval_active_learning.py
(same split scheme with random_state=model_id, same selection families,
same /assets/active_learning pickle layout)."""

import logging
import pickle
from typing import Callable, Dict, List, Optional, Tuple

import numpy as np
import torch
from sklearn.model_selection import train_test_split

from ..config import assets_path, ensure_dir
from .coverage_handler import CoverageWorker
from .model_handler import BaseModel
from .surprise_handler import SurpriseHandler

logger = logging.getLogger(__name__)

RANDOM_SPLIT = "random"
NOM, OOD = "nominal", "ood"
OBS, FUT = "observed", "future"

SplitDataset = Dict[Tuple[str, str], Tuple[np.ndarray, np.ndarray]]
SplitEvaluation = Dict[Tuple[str, str], float]
MetricSelection = Dict[Tuple[str, str], List[int]]


def evaluate(
    model_id: int,
    case_study: str,
    model,
    train_x,
    train_y,
    nominal_test_x,
    nominal_test_labels,
    ood_test_x,
    ood_test_labels,
    nc_activation_layers: List[int],
    sa_activation_layers: List[int],
    training_process: Callable,
    observed_share: float,
    num_selected: int,
    num_classes: Optional[int],
    dsa_badge_size: Optional[int] = None,
    device=None,
    predict_batch: int = 512,
) -> None:
    """Evaluate active-learning value of every TIP for one model id."""
    datasets = _shuffle_and_split_datasets(
        model_id, nominal_test_x, nominal_test_labels, ood_test_x, ood_test_labels,
        observed_share,
    )
    original_model_eval = _evaluate_model(model, datasets, device)

    selections: MetricSelection = {}
    selections.update(_get_fp_selection(model, datasets, num_selected, device, predict_batch))
    selections.update(
        _get_nc_selection(model, train_x, datasets, nc_activation_layers, num_selected, device, predict_batch)
    )
    selections.update(
        _get_sa_selection(
            model, train_x, datasets, sa_activation_layers, num_selected,
            dsa_badge_size, device, predict_batch,
        )
    )
    selections.update(_get_random_selection(datasets, num_selected))
    _selection_sanity_checks(num_selected, selections)

    active_accuracies = {}
    for (metric, ood_or_nom), selected in selections.items():
        sel = np.asarray(selected)
        x = datasets[ood_or_nom, OBS][0][sel]
        y = datasets[ood_or_nom, OBS][1][sel]
        new_model = _retrain(training_process, train_x, train_y, x, y)
        active_accuracies[(metric, ood_or_nom)] = _evaluate_model(new_model, datasets, device)
        del new_model

    _save_results(case_study, model_id, "original", "na", original_model_eval)
    for (metric, ood_or_nom), eval_res in active_accuracies.items():
        _save_results(case_study, model_id, metric, ood_or_nom, eval_res)


def _save_results(case_study, model_id, metric, ood_or_nom, eval_res: SplitEvaluation):
    ensure_dir("active_learning")
    path = assets_path(
        "active_learning", f"{case_study}_{model_id}_{metric}_{ood_or_nom}.pickle"
    )
    with open(path, "wb") as f:
        pickle.dump(eval_res, f)


def _selection_sanity_checks(num_selected, selections):
    for (metric, ood_or_nom), sel in selections.items():
        assert len(sel) == num_selected, (
            f"selection size for {metric}, {ood_or_nom}: {len(sel)} != {num_selected}"
        )
        assert len(set(int(i) for i in sel)) == num_selected, (
            f"selection for {metric}, {ood_or_nom} is not unique"
        )


def _retrain(training_process, train_x, train_y, new_x, new_y):
    """Retrain from scratch on train + selected (reference semantics:
    concat, shuffle, full re-train — eval_active_learning.py:161-180)."""
    x = np.concatenate((np.asarray(train_x), np.asarray(new_x)))
    y = np.concatenate(
        (np.asarray(train_y).reshape(-1), np.asarray(new_y).reshape(-1))
    )
    idx = np.random.permutation(len(x))
    return training_process(x[idx], y[idx])


def _get_random_selection(datasets: SplitDataset, num_selected: int) -> MetricSelection:
    res: MetricSelection = {}
    for (ood_or_nom, obs_or_fut), _ in datasets.items():
        if obs_or_fut == OBS:
            res[RANDOM_SPLIT, ood_or_nom] = list(range(num_selected))
    return res


def _get_fp_selection(model, datasets, num_selected, device, predict_batch) -> MetricSelection:
    res: MetricSelection = {}
    base_model = BaseModel(model, activation_layers=None, device=device, predict_batch=predict_batch)
    for (ood_or_nom, obs_or_fut), (x, y) in datasets.items():
        if obs_or_fut != OBS:
            continue
        _, uncertainties, _ = base_model.get_pred_and_uncertainty(x)
        for metric, unc in uncertainties.items():
            res[metric, ood_or_nom] = np.argsort(unc)[-num_selected:].tolist()
    return res


def _get_nc_selection(model, train_x, datasets, nc_layers, num_selected, device, predict_batch) -> MetricSelection:
    res: MetricSelection = {}
    worker = CoverageWorker(
        base_model=BaseModel(model, activation_layers=nc_layers, device=device, predict_batch=predict_batch),
        training_set=train_x,
    )
    for (ood_or_nom, obs_or_fut), (x, y) in datasets.items():
        if obs_or_fut != OBS:
            continue
        _, all_scores, cam_orders = worker.evaluate_all(x, f"al-{ood_or_nom}")
        for metric, scores in all_scores.items():
            res[metric, ood_or_nom] = np.argsort(scores, kind="stable")[-num_selected:].tolist()
        for metric, cam_order in cam_orders.items():
            res[f"{metric}-cam", ood_or_nom] = list(cam_order[:num_selected])
    return res


def _get_sa_selection(
    model, train_x, datasets, sa_layers, num_selected, dsa_badge_size, device, predict_batch
) -> MetricSelection:
    res: MetricSelection = {}
    worker = SurpriseHandler(
        model=model, sa_layers=sa_layers, training_dataset=train_x,
        device=device, predict_batch=predict_batch,
    )
    results = worker.evaluate_all(
        datasets={NOM: datasets[NOM, OBS][0], OOD: datasets[OOD, OBS][0]},
        dsa_badge_size=dsa_badge_size,
    )
    for metric, values in results.items():
        for nom_or_ood, (sa, cam_order, _) in values.items():
            res[metric, nom_or_ood] = np.argsort(sa, kind="stable")[-num_selected:].tolist()
            res[f"{metric}-cam", nom_or_ood] = list(cam_order[:num_selected])
    return res


def _shuffle_and_split_datasets(
    model_id, nominal_x, nominal_y, ood_x, ood_y, observed_share
) -> SplitDataset:
    res: SplitDataset = {}
    fut_x, obs_x, fut_y, obs_y = train_test_split(
        np.asarray(nominal_x), np.asarray(nominal_y), test_size=observed_share,
        random_state=model_id,
    )
    res[NOM, OBS] = (obs_x, obs_y)
    res[NOM, FUT] = (fut_x, fut_y)
    fut_x, obs_x, fut_y, obs_y = train_test_split(
        np.asarray(ood_x), np.asarray(ood_y), test_size=observed_share,
        random_state=model_id,
    )
    res[OOD, OBS] = (obs_x, obs_y)
    res[OOD, FUT] = (fut_x, fut_y)
    return res


@torch.no_grad()
def _evaluate_model(model, datasets: SplitDataset, device, batch: int = 512) -> SplitEvaluation:
    """Accuracy of the model on all four splits."""
    model.eval()
    dev = device or next(model.parameters()).device
    res: SplitEvaluation = {}
    for (ood_or_nom, obs_or_fut), (x, y) in datasets.items():
        correct = 0
        xt = torch.as_tensor(np.ascontiguousarray(x))
        if xt.dtype == torch.float64:
            xt = xt.float()
        yt = torch.as_tensor(np.asarray(y).reshape(-1))
        for s in range(0, xt.shape[0], batch):
            logits = model(xt[s : s + batch].to(dev))
            correct += int((logits.argmax(dim=1).cpu() == yt[s : s + batch]).sum())
        acc = correct / xt.shape[0]
        assert 0 <= acc <= 1
        res[ood_or_nom, obs_or_fut] = acc
    return res
