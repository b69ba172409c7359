import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu on a GPU box)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def small_config():
    from improved_body_parts_amd.config import CanonicalConfig
    return CanonicalConfig(128, 128, 4)


@pytest.fixture(scope="session")
def small_opt():
    from improved_body_parts_amd.config import TrainingOpt
    return TrainingOpt(nstack=2, hourglass_inp_dim=64, increase=32, batch_size=2,
                       nstack_weight=[1, 1])
