import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU")


@pytest.fixture(scope="session")
def device():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    return torch.device("cuda:0")
