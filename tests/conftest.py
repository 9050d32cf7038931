import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a ROCm GPU (run on MI355X via gpurun)"
    )


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def rng():
    return np.random.default_rng(42)


@pytest.fixture
def small_blobs():
    from cuda_gmm_mpi_amd.utils.synthetic import make_blobs
    data, labels = make_blobs(2000, 3, 4, seed=7)
    return data, labels
