import os

import pytest
import torch

# run the suites with the numeric guards ON (they are opt-in in
# production because each check synchronizes the device)
os.environ.setdefault("CODA_AMD_DEBUG", "1")


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a ROCm GPU (run on an MI355X box)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture(scope="session")
def synthetic_task():
    from coda_amd.datasets import make_synthetic_task
    preds, labels = make_synthetic_task(H=8, N=300, C=5, seed=0)
    return preds, labels


@pytest.fixture()
def synthetic_dataset(synthetic_task):
    from coda_amd.datasets import Dataset
    preds, labels = synthetic_task
    return Dataset.from_tensors(preds, labels, "cpu")
