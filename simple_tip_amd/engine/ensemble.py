"""Process-per-model ensemble runner.

Replaces uncertainty-wizard's ``LazyEnsemble`` (the reference's only
parallelism: case_study.py:19-25, 87-144) with a plain multiprocessing
spawn pool — one task per process (the reference's SingleUseContext
semantics, memory_leak_avoider.py:13-23), torch checkpoints under
/assets/models/{study}/{id}.pt.
"""

import logging
import multiprocessing as mp
import os
from typing import Any, Callable, List, Optional

import torch

from ..config import ensure_dir, assets_path

logger = logging.getLogger(__name__)


def model_path(study_name: str, model_id: int) -> str:
    ensure_dir("models", study_name)
    return assets_path("models", study_name, f"{model_id}.pt")


def save_model(study_name: str, model_id: int, model: torch.nn.Module) -> None:
    torch.save(
        {"state_dict": model.state_dict()}, model_path(study_name, model_id)
    )


def load_model(study_name: str, model_id: int, factory: Callable[[], torch.nn.Module], device=None) -> torch.nn.Module:
    model = factory()
    ckpt = torch.load(model_path(study_name, model_id), map_location="cpu", weights_only=True)
    model.load_state_dict(ckpt["state_dict"])
    if device is not None:
        model = model.to(device)
    model.eval()
    return model


def _run_on_assigned_gpu(task: Callable[[int], Any], ngpu: int, mid: int):
    """Pool target: pin this (fresh, spawned) worker to GPU ``mid % ngpu``
    BEFORE torch initialises HIP — model ids spread round-robin over the
    node's GPUs, which is the natural placement for the reference's
    "100 independent models" experiment shape on an 8-GPU node."""
    if ngpu > 1 and "HIP_VISIBLE_DEVICES" not in os.environ:
        os.environ["HIP_VISIBLE_DEVICES"] = str(mid % ngpu)
    return task(mid)


def run_tasks(
    task: Callable[[int], Any],
    model_ids: List[int],
    num_processes: int = 0,
) -> List[Any]:
    """Run ``task(model_id)`` for each id.

    num_processes == 0 runs inline (tests, single-GPU boxes); otherwise a
    spawn pool with maxtasksperchild=1 (one task per process, then exit —
    the reference's TF memory-leak workaround, kept because each child also
    gets a fresh HIP context). On a multi-GPU node each model id is pinned
    to GPU ``id % num_gpus`` (round-robin model-level parallelism).
    ``task`` must be picklable (module-level function / functools.partial).
    """
    if num_processes <= 0 or len(model_ids) <= 1:
        return [task(mid) for mid in model_ids]
    from functools import partial

    ngpu = torch.cuda.device_count() if torch.cuda.is_available() else 0
    ctx = mp.get_context("spawn")
    with ctx.Pool(processes=num_processes, maxtasksperchild=1) as pool:
        return pool.map(partial(_run_on_assigned_gpu, task, ngpu), model_ids)
