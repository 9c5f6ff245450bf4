import os
import sys

import pytest
import torch

# repo root on sys.path so `run_experiment` and `turboprune_amd` import
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X (ROCm) GPU")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def tiny_cifar_cfg(tmp_path):
    """Minimal CIFAR10 config: tiny synthetic data, 1 epoch."""
    from turboprune_amd.config import compose
    return compose("cifar10_er_erk", [
        "experiment_params.epochs_per_level=1",
        "dataset_params.total_batch_size=64",
        "+dataset_params.synthetic_size=256",
        f"experiment_params.base_dir={tmp_path}/experiments",
        f"dataset_params.data_root_dir={tmp_path}/data",
        "pruning_params.target_sparsity=0.9",
    ])


@pytest.fixture
def device():
    return torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
