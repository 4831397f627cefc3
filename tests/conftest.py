import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X GPU (run via gpurun)")
    # starlette's TestClient warns about its own httpx dependency —
    # environment-level noise, not ours
    config.addinivalue_line(
        "filterwarnings",
        "ignore::starlette.exceptions.StarletteDeprecationWarning")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def cpu_cfg():
    from scaffold.config import load_config

    return load_config(
        None,
        [
            "model.name=mlp",
            "data.kind=synthetic_vector",
            "data.batch_size=32",
            "dtype=fp32",
            "device=cpu",
            "steps=20",
            "log.interval=1000",
        ],
    )

