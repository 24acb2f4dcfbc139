import os

import pytest
import torch

# The suite runs on synthetic data with random-init weights; the hash
# tokenizer is the sanctioned stand-in (the hygiene test for the
# non-opt-in error path clears this var via monkeypatch).
os.environ.setdefault("DISTLLM_ALLOW_HASH_TOKENIZER", "1")


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU")


@pytest.fixture(scope="session")
def device():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    return torch.device("cuda:0")
