import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
if str(REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(REPO_ROOT))

import gpustack_amd.engine  # noqa: F401,E402  (import order: engine before
# models — a test importing gpustack_amd.models first would hit the
# llama<->model_runner package cycle)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    import torch

    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
