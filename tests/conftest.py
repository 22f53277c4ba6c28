import os
import sys

import pytest
import torch

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a ROCm GPU (run on the MI355X box)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def tiny_mnist_blob():
    from tools.create_data import make_mnist_blob
    return make_mnist_blob(n_users=20, samples_per_user=12, seed=3)


@pytest.fixture
def tmp_data_dir(tmp_path):
    """Tiny synthetic MNIST dataset tree for e2e runs."""
    from tools.create_data import make_mnist_blob, save_blob
    d = tmp_path / "data"
    save_blob(make_mnist_blob(30, 15, seed=0), str(d / "cv_lr_mnist" / "train_data.pt"))
    save_blob(make_mnist_blob(4, 15, seed=1), str(d / "cv_lr_mnist" / "val_data.pt"))
    save_blob(make_mnist_blob(4, 15, seed=2), str(d / "cv_lr_mnist" / "test_data.pt"))
    return str(d)
