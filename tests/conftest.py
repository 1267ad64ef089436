import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a visible MI355X GPU (run on the GPU box)")


@pytest.fixture(scope="session")
def gpu_available():
    try:
        import torch
        return torch.cuda.is_available()
    except Exception:
        return False
