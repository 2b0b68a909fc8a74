import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def rng():
    return np.random.default_rng(0)


@pytest.fixture(autouse=True)
def _gpu_test_hygiene():
    """Deterministic teardown between GPU tests: destroy dead CUDAGraph /
    allocator state NOW (GC'ing a CUDAGraph mid-capture of a later test
    intermittently faults on this ROCm build) and drain the device."""
    yield
    if torch.cuda.is_available():
        import gc

        torch.cuda.synchronize()
        gc.collect()
        torch.cuda.synchronize()
