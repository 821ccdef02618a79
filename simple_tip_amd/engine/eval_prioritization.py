"""Test-prioritization experiment: scores, CAM orders, misclassification
masks and timing artifacts for one (model, case study).

Capability parity with reference src/dnn_test_prio/eval_prioritization.py
(same /assets artifact names: priorities/{cs}_{ds}_{id}_{type}.npy and
times/{cs}_{ds}_{id}_{metric} pickles)."""

import logging
import pickle
from typing import Dict, List, Optional

import numpy as np

from ..config import assets_path, ensure_dir
from .coverage_handler import CoverageWorker
from .model_handler import BaseModel
from .surprise_handler import SurpriseHandler

logger = logging.getLogger(__name__)


def _is_rank0() -> bool:
    from ..parallel.dist import get_rank

    return get_rank() == 0


def _persist(case_study: str, dataset_id: str, data_type: str, model_id: int, data):
    if not _is_rank0():
        return  # ranks hold identical artifacts; one writer avoids races
    ensure_dir("priorities")
    np.save(
        assets_path(
            "priorities", f"{case_study}_{dataset_id}_{model_id}_{data_type}.npy"
        ),
        np.asarray(data),
    )


def _persist_times(case_study: str, dataset_id: str, model_id: int, metric: str, data: List[float]):
    if not _is_rank0():
        return
    ensure_dir("times")
    with open(
        assets_path("times", f"{case_study}_{dataset_id}_{model_id}_{metric}"), "wb"
    ) as f:
        pickle.dump(data, f)


def _persist_times_multiple_metrics(case_study, dataset_id, model_id, data: Dict[str, List[float]]):
    # written per metric so partial re-runs lose nothing
    for metric, times in data.items():
        _persist_times(case_study, dataset_id, model_id, metric, times)


def load(case_study: str, dataset_id: str, data_type: str, model_id: int) -> np.ndarray:
    """Load a priorities artifact."""
    return np.load(
        assets_path(
            "priorities", f"{case_study}_{dataset_id}_{model_id}_{data_type}.npy"
        )
    )


def evaluate(
    model_id: int,
    case_study: str,
    model,
    training_dataset,
    nominal_test_dataset,
    nominal_test_labels,
    ood_test_dataset,
    ood_test_labels,
    nc_activation_layers: List[int],
    sa_activation_layers: List[int],
    dsa_badge_size: Optional[int] = None,
    device=None,
    predict_batch: int = 512,
    dist_shard: bool = False,
) -> None:
    """Run all TIP families for one model and persist every artifact.

    ``dist_shard=True`` on an initialised torch.distributed group runs the
    whole experiment data-parallel over the test-input axis (RCCL on GPU,
    gloo on CPU): every rank forwards/scores its contiguous input shard,
    per-input scores all-gather, coverage profiles reassemble with the
    bitmap OR all-reduce, and CAM runs identically on every rank from the
    full matrices. Artifacts are written by rank 0 and are identical to a
    single-process run (same batch content per forward).
    """
    _eval_fault_predictors(
        case_study, model, model_id, nominal_test_dataset, nominal_test_labels,
        "nominal", device, predict_batch, dist_shard,
    )
    _eval_fault_predictors(
        case_study, model, model_id, ood_test_dataset, ood_test_labels, "ood",
        device, predict_batch, dist_shard,
    )
    _eval_neuron_coverage(
        case_study, model, model_id, nc_activation_layers,
        nominal_test_dataset, ood_test_dataset, training_dataset, device,
        predict_batch, dist_shard,
    )
    _eval_surprise(
        case_study, model, model_id, sa_activation_layers,
        nominal_test_dataset, ood_test_dataset, training_dataset,
        dsa_badge_size, device, predict_batch, dist_shard,
    )


def _eval_fault_predictors(
    case_study, model, model_id, ds, labels, ds_type, device, predict_batch,
    dist_shard=False,
):
    from ..parallel.dist import allgather_rows, get_world_size, shard_slice

    base_model = BaseModel(model, activation_layers=None, device=device, predict_batch=predict_batch)
    if dist_shard and get_world_size() > 1:
        import torch

        n = ds.shape[0]
        pred_l, unc_l, times = base_model.get_pred_and_uncertainty(
            ds[shard_slice(n)]
        )
        pred = allgather_rows(torch.from_numpy(pred_l), n).numpy()
        uncertainties = {
            k: allgather_rows(torch.from_numpy(v), n).numpy()
            for k, v in unc_l.items()
        }
    else:
        pred, uncertainties, times = base_model.get_pred_and_uncertainty(ds)
    is_misclassified = pred != np.asarray(labels).reshape(-1)
    _persist(case_study, ds_type, "is_misclassified", model_id, is_misclassified)
    _persist_times_multiple_metrics(case_study, ds_type, model_id, times)
    for unc_id, unc in uncertainties.items():
        _persist(case_study, ds_type, f"uncertainty_{unc_id}", model_id, unc)


def _eval_neuron_coverage(
    case_study, model, model_id, layers, nominal_test_dataset, ood_test_dataset,
    training_dataset, device, predict_batch, dist_shard=False,
):
    nc_worker = CoverageWorker(
        base_model=BaseModel(model, activation_layers=layers, device=device, predict_batch=predict_batch),
        training_set=training_dataset,
        dist_shard=dist_shard,
    )
    for name, ds in {"nominal": nominal_test_dataset, "ood": ood_test_dataset}.items():
        times, scores, cam_orders = nc_worker.evaluate_all(ds, name)
        _persist_times_multiple_metrics(case_study, name, model_id, times)
        for metric_id, score in scores.items():
            _persist(case_study, name, f"{metric_id}_scores", model_id, score)
        for metric_id, order in cam_orders.items():
            _persist(case_study, name, f"{metric_id}_cam_order", model_id, np.array(order))


def _eval_surprise(
    case_study, model, model_id, layers, nominal_test_dataset, ood_test_dataset,
    training_dataset, dsa_badge_size, device, predict_batch, dist_shard=False,
):
    sa_worker = SurpriseHandler(
        model=model, sa_layers=layers, training_dataset=training_dataset,
        device=device, predict_batch=predict_batch, dist_shard=dist_shard,
    )
    results = sa_worker.evaluate_all(
        datasets={"nominal": nominal_test_dataset, "ood": ood_test_dataset},
        dsa_badge_size=dsa_badge_size,
    )
    for metric, values in results.items():
        for dataset, (sa, cam_order, times) in values.items():
            _persist_times(case_study, dataset, model_id, metric, times)
            _persist(case_study, dataset, f"{metric}_scores", model_id, sa)
            _persist(case_study, dataset, f"{metric}_cam_order", model_id, cam_order)
