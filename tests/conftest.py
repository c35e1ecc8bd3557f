import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a ROCm GPU (run with -m gpu on an MI355X box)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def tiny_config():
    from mine_amd.config import default_config
    return default_config(**{
        "data.name": "synthetic",
        "data.img_h": 96,
        "data.img_w": 128,
        "mpi.num_bins_coarse": 8,
        "data.per_gpu_batch_size": 2,
        "data.visible_point_count": 32,
        "training.amp_dtype": "fp32",
        "data.num_workers": 0,
    })
