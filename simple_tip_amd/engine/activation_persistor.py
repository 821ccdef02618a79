"""Persist per-layer activation traces to the assets fabric.

Capability parity with reference src/dnn_test_prio/activation_persistor.py
(layout: /assets/activations/{cs}/model_{id}/{ds}/layer_{i}/badge_{b}.npy,
BADGE_SIZE=100)."""

import os
from typing import Dict, List

import numpy as np

from .. import config
from .model_handler import BaseModel, iter_batches

BADGE_SIZE = 100


def persist(
    case_study: str,
    model_id: int,
    model,
    datasets: Dict[str, object],
    num_layers: int,
    device=None,
) -> None:
    """Dump every layer's activations per badge for each dataset."""
    base_model = BaseModel(
        model,
        activation_layers=list(range(num_layers)),
        device=device,
        predict_batch=BADGE_SIZE,
    )
    for ds_name, data in datasets.items():
        for b, acts in enumerate(
            base_model.walk_activations(iter_batches(data, BADGE_SIZE))
        ):
            for i, layer in enumerate(acts):
                folder = os.path.join(
                    config.OUTPUT_FOLDER,
                    "activations",
                    case_study,
                    f"model_{model_id}",
                    ds_name,
                    f"layer_{i}",
                )
                os.makedirs(folder, exist_ok=True)
                np.save(
                    os.path.join(folder, f"badge_{b}.npy"),
                    layer.cpu().numpy(),
                )
