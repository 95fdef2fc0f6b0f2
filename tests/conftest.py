import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a ROCm GPU (run on MI355X via gpurun)")
    config.addinivalue_line("markers", "dist: multi-process test (gloo on CPU)")


@pytest.fixture(autouse=True)
def _seed():
    torch.manual_seed(0)


def requires_gpu():
    return pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")
