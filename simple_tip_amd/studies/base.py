"""Abstract case study: owns data, model factory, training loop and the four
experiment phases (train / test_prio / active_learning / at_collection).

Capability parity with reference src/dnn_test_prio/case_study.py:13-144.
The ensemble is a spawn process pool over model ids (engine/ensemble.py);
training runs in bf16 autocast on the GPU.
"""

import logging
import os
from functools import partial
from typing import List, Optional, Tuple

import numpy as np
import torch
import torch.nn as nn

from ..config import MAX_NUM_MODELS, StudyConfig
from ..engine import ensemble
from ..engine import eval_prioritization, eval_active_learning, activation_persistor
from ..models.base import TapModel

logger = logging.getLogger(__name__)


def default_device() -> torch.device:
    return torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")


def train_classifier(
    model: TapModel,
    x: np.ndarray,
    y: np.ndarray,
    epochs: int,
    batch_size: int,
    lr: float = 1e-3,
    device: Optional[torch.device] = None,
    validation_split: float = 0.1,
    seed: Optional[int] = None,
) -> TapModel:
    """Adam + cross-entropy training loop (bf16 autocast on GPU).

    Plays the role of the reference's keras ``model.fit`` calls
    (e.g. case_study_mnist.py:68)."""
    device = device or default_device()
    if seed is not None:
        torch.manual_seed(seed)
    model = model.to(device)
    model.train()

    # Data-parallel training over RCCL/xGMI when torch.distributed is up
    # (the 8-GPU active-learning retrain path). The models are <1M params,
    # so DDP uses a single gradient bucket — one small all-reduce per step.
    from ..parallel import dist as pdist

    train_model = model
    world = pdist.get_world_size()
    if world > 1:
        from torch.nn.parallel import DistributedDataParallel as DDP

        ids = [device.index] if device.type == "cuda" else None
        train_model = DDP(model, device_ids=ids, bucket_cap_mb=32)

    opt = torch.optim.Adam(train_model.parameters(), lr=lr)
    loss_fn = nn.CrossEntropyLoss()
    n = x.shape[0]
    n_train = int(n * (1 - validation_split)) if validation_split else n
    xt = torch.as_tensor(np.ascontiguousarray(x))
    if xt.dtype == torch.float64:
        xt = xt.float()
    yt = torch.as_tensor(np.asarray(y).reshape(-1)).long()
    use_amp = device.type == "cuda"
    # NHWC training on GPU for conv models: bf16-autocast NCHW convs hit
    # MIOpen's NAIVE weight-gradient fallback (~30 ms/launch measured in
    # bench traces); channels_last routes the backward to the igemm path.
    nhwc = use_amp and xt.dim() == 4
    if nhwc:
        model.to(memory_format=torch.channels_last)
    if use_amp:
        # keep the whole training set resident in HBM (reference sets are
        # <= 750 MB of the 288 GB): per-step batches become device gathers
        # instead of CPU indexing + H2D copies — at the reference's small
        # batch sizes (cifar10: 32) the host side dominated the epoch
        xt = xt.to(device)
        yt = yt.to(device)
    for epoch in range(epochs):
        # identical permutation on every rank, rank-strided shard of it
        gen = torch.Generator().manual_seed((seed or 0) * 1000 + epoch)
        perm = torch.randperm(n_train, generator=gen)
        rank = pdist.get_rank()
        total = 0
        # epoch statistics accumulate ON DEVICE: a float()/int() per step is
        # a host sync, and at reference batch sizes (cifar10: batch 32,
        # 1.5k steps/epoch) the syncs dominated the epoch wall time
        loss_acc = torch.zeros((), dtype=torch.float64, device=device)
        corr_acc = torch.zeros((), dtype=torch.int64, device=device)
        total_steps = max(1, (n_train + batch_size - 1) // batch_size)
        # every rank runs the same number of steps (DDP all-reduce must
        # match); overhanging ranks wrap to batch 0
        for t in range((total_steps + world - 1) // world):
            k = (t * world + rank) % total_steps
            idx = perm[k * batch_size : (k + 1) * batch_size]
            if idx.numel() == 0:
                idx = perm[0:batch_size]
            idx = idx.to(xt.device)
            xb = xt[idx].to(device, non_blocking=True)
            yb = yt[idx].to(device, non_blocking=True)
            if nhwc:
                xb = xb.to(memory_format=torch.channels_last)
            opt.zero_grad(set_to_none=True)
            with torch.autocast("cuda", dtype=torch.bfloat16, enabled=use_amp):
                logits = train_model(xb)
                loss = loss_fn(logits.float(), yb)
            loss.backward()
            opt.step()
            loss_acc += loss.detach().double() * len(idx)
            corr_acc += (logits.argmax(dim=1) == yb).sum()
            total += len(idx)
        logger.info(
            "epoch %d/%d loss %.4f acc %.3f", epoch + 1, epochs,
            float(loss_acc) / max(total, 1), int(corr_acc) / max(total, 1),
        )
    model.eval()
    return model


class CaseStudy:
    """Base class; subclasses provide config, model factory and data."""

    config: StudyConfig

    def __init__(self, scale: Optional[float] = None, device=None):
        # scale < 1 shrinks dataset sizes/epochs (CPU tests, smoke runs)
        self.scale = scale if scale is not None else float(
            os.environ.get("TIP_SCALE", "1.0")
        )
        self.device = device or default_device()

    # -- to be provided by subclasses -----------------------------------
    def build_model(self) -> TapModel:
        raise NotImplementedError

    def load_datasets(self) -> Tuple[
        Tuple[np.ndarray, np.ndarray],
        Tuple[np.ndarray, np.ndarray],
        Tuple[np.ndarray, np.ndarray],
    ]:
        """((train_x, train_y), (nominal_x, nominal_y), (ood_x, ood_y))."""
        raise NotImplementedError

    # -- derived sizes ---------------------------------------------------
    def _n(self, n: int) -> int:
        return max(64, int(n * self.scale))

    def _epochs(self) -> int:
        return max(1, int(round(self.config.epochs * min(1.0, self.scale * 2))))

    def training_process(self, x: np.ndarray, y: np.ndarray) -> TapModel:
        """Train a fresh model on (x, y) — used by train and retrain."""
        return train_classifier(
            self.build_model(),
            x,
            y,
            epochs=self._epochs(),
            batch_size=self.config.train_batch,
            device=self.device,
        )

    # -- phases ----------------------------------------------------------
    def train(self, model_ids: List[int], num_processes: int = 0, context=None) -> None:
        """Train and persist one model per id."""
        assert all(0 <= i < MAX_NUM_MODELS for i in model_ids)
        ensemble.run_tasks(
            partial(_train_task, type(self), self.scale), model_ids, num_processes
        )

    def run_prio_eval(self, model_ids: List[int], num_processes: int = 0, context=None) -> None:
        ensemble.run_tasks(
            partial(_prio_task, type(self), self.scale), model_ids, num_processes
        )

    def run_active_learning_eval(
        self, model_ids: List[int], num_processes: int = 0, context=None
    ) -> None:
        ensemble.run_tasks(
            partial(_active_task, type(self), self.scale), model_ids, num_processes
        )

    def collect_activations(
        self, model_ids: List[int], num_processes: int = 0, context=None
    ) -> None:
        ensemble.run_tasks(
            partial(_at_collection_task, type(self), self.scale), model_ids, num_processes
        )

    # -- helpers ---------------------------------------------------------
    def _load_model(self, model_id: int) -> TapModel:
        return ensemble.load_model(
            self.config.name, model_id, self.build_model, device=self.device
        )


# -- module-level task functions (picklable for the spawn pool) -------------

def _train_task(study_cls, scale, model_id: int):
    study = study_cls(scale=scale)
    (train_x, train_y), _, _ = study.load_datasets()
    torch.manual_seed(model_id)
    np.random.seed(model_id)
    model = study.training_process(train_x, train_y)
    ensemble.save_model(study.config.name, model_id, model)
    logger.info("trained and saved %s model %d", study.config.name, model_id)


def _prio_task(study_cls, scale, model_id: int):
    study = study_cls(scale=scale)
    (train_x, _), (nom_x, nom_y), (ood_x, ood_y) = study.load_datasets()
    model = study._load_model(model_id)
    eval_prioritization.evaluate(
        model_id=model_id,
        case_study=study.config.name,
        model=model,
        training_dataset=train_x,
        nominal_test_dataset=nom_x,
        nominal_test_labels=nom_y,
        ood_test_dataset=ood_x,
        ood_test_labels=ood_y,
        nc_activation_layers=study.config.nc_layers,
        sa_activation_layers=study.config.sa_layers,
        dsa_badge_size=study.config.dsa_badge_size,
        device=study.device,
        predict_batch=study.config.predict_batch,
    )


def _active_task(study_cls, scale, model_id: int):
    study = study_cls(scale=scale)
    (train_x, train_y), (nom_x, nom_y), (ood_x, ood_y) = study.load_datasets()
    model = study._load_model(model_id)
    eval_active_learning.evaluate(
        model_id=model_id,
        case_study=study.config.name,
        model=model,
        train_x=train_x,
        train_y=train_y,
        nominal_test_x=nom_x,
        nominal_test_labels=nom_y,
        ood_test_x=ood_x,
        ood_test_labels=ood_y,
        nc_activation_layers=study.config.nc_layers,
        sa_activation_layers=study.config.sa_layers,
        training_process=study.training_process,
        observed_share=study.config.observed_share,
        num_selected=max(8, int(study.config.num_selected * study.scale)),
        num_classes=study.config.num_classes,
        dsa_badge_size=study.config.dsa_badge_size,
        device=study.device,
        predict_batch=study.config.predict_batch,
    )


def _at_collection_task(study_cls, scale, model_id: int):
    study = study_cls(scale=scale)
    (train_x, _), (nom_x, _), (ood_x, _) = study.load_datasets()
    model = study._load_model(model_id)
    activation_persistor.persist(
        case_study=study.config.name,
        model_id=model_id,
        model=model,
        datasets={"train": train_x, "nominal": nom_x, "ood": ood_x},
        num_layers=len(model.layers),
        device=study.device,
    )
