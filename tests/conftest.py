import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X GPU (run via gpurun)")


@pytest.fixture(scope="session")
def gpu_available():
    import torch

    return torch.cuda.is_available()
