import gc

import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests that require an MI355X GPU")
    config.addinivalue_line("markers", "dist: multi-process distributed tests (gloo on CPU)")
    config.addinivalue_line("markers", "largedist: tests that need 8 GPUs")


@pytest.fixture(autouse=True)
def clean_cache():
    yield
    gc.collect()
    if torch.cuda.is_available():
        torch.cuda.empty_cache()
