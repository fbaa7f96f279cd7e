import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a ROCm GPU (run on MI355X via gpurun)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def ae_config():
    from dsin_amd import config as cm
    cfg, _ = cm.parse(os.path.join(os.path.dirname(__file__), "..",
                                   "run_configs", "ae_run_configs"))
    return cfg


@pytest.fixture
def pc_config():
    from dsin_amd import config as cm
    cfg, _ = cm.parse(os.path.join(os.path.dirname(__file__), "..",
                                   "run_configs", "pc_run_configs"))
    return cfg


@pytest.fixture
def small_ae_config(ae_config):
    cfg = ae_config.clone()
    cfg.crop_size = (64, 96)
    cfg.y_patch_size = (16, 16)
    return cfg
