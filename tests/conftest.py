import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires a ROCm GPU (MI355X)")


@pytest.fixture(scope="session")
def gpu_device():
    import torch

    if not torch.cuda.is_available():
        pytest.skip("no GPU available")
    return torch.device("cuda:0")
