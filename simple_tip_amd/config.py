"""Global configuration: artifact layout and experiment-scale constants.

Mirrors the reference's module-level constants (reference:
src/dnn_test_prio/case_study.py:9-10, handler_model.py:7,
handler_surprise.py:14) but centralised and overridable via environment
variables instead of being scattered per-module.
"""

import os
from dataclasses import dataclass, field
from typing import List, Optional


def _default_assets_root() -> str:
    env = os.environ.get("TIP_ASSETS_DIR")
    if env:
        return env
    # The reference hard-codes "/assets/" (case_study.py:10). Keep that when it
    # is present (docker-style mount), fall back to a repo-local folder.
    if os.path.isdir("/assets") and os.access("/assets", os.W_OK):
        return "/assets"
    return os.path.abspath("./assets")


#: Root of the artifact fabric (models, priorities, times, results, ...)
OUTPUT_FOLDER = _default_assets_root()

#: Max models per case study (reference: case_study.py:9)
MAX_NUM_MODELS = 100

#: MC-dropout samples for the variation-ratio quantifier
#: (reference: handler_model.py:7)
DROPOUT_SAMPLE_SIZE = 200

#: Number of surprise-coverage buckets (reference: handler_surprise.py:14)
NUM_SC_BUCKETS = 1000


def assets_path(*parts: str, create_parent: bool = False) -> str:
    """Join a path under the assets root, optionally creating the directory."""
    p = os.path.join(OUTPUT_FOLDER, *parts)
    if create_parent:
        os.makedirs(os.path.dirname(p), exist_ok=True)
    return p


def ensure_dir(*parts: str) -> str:
    """Create (if needed) and return a directory under the assets root."""
    p = os.path.join(OUTPUT_FOLDER, *parts)
    os.makedirs(p, exist_ok=True)
    return p


@dataclass
class StudyConfig:
    """Typed per-case-study configuration.

    Replaces the reference's per-module constants
    (e.g. case_study_mnist.py:25-29, case_study_imdb.py:23-43).
    """

    name: str
    num_classes: int
    input_shape: tuple  # CHW for images, (seq_len,) for text
    train_size: int
    test_size: int
    # Activation-tap layer indices (indices into the model's layer list)
    sa_layers: List[int] = field(default_factory=list)
    nc_layers: List[int] = field(default_factory=list)
    # Training hyper-parameters
    epochs: int = 10
    train_batch: int = 128
    # Prediction batch size (reference: 32 default, IMDB 600 via
    # model.custom_badge_size — handler_model.py:126-130)
    predict_batch: int = 512
    # Active learning
    observed_share: float = 0.5
    num_selected: int = 1000
    # DSA badge override (reference: case_study_imdb.py:254)
    dsa_badge_size: Optional[int] = None
