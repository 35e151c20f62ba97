import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on MI355X via gpurun)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(autouse=True)
def _fresh_registry():
    from deeprec_amd.embedding.variable import reset_registry, GLOBAL_STEP
    reset_registry()
    GLOBAL_STEP.value = 0
    yield
    reset_registry()
    GLOBAL_STEP.value = 0
