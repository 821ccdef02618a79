import os
import tempfile

# Route the /assets artifact fabric to a scratch dir BEFORE any
# simple_tip_amd import resolves the layout.
os.environ.setdefault(
    "TIP_ASSETS_DIR", tempfile.mkdtemp(prefix="tip_assets_test_")
)

import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (MI355X) and the HIP extension"
    )


def pytest_collection_modifyitems(config, items):
    import torch

    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
