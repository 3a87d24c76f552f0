import random

import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests that require a ROCm GPU (run on MI355X via gpurun)")


def pytest_sessionstart(session):
    # determinism across the suite (mirrors the reference's conftest seed discipline)
    random.seed(1)
    np.random.seed(1)
    torch.manual_seed(1)


@pytest.fixture
def device():
    return "cuda" if torch.cuda.is_available() else "cpu"


try:  # deterministic hypothesis examples: CI runs must not flake
    from hypothesis import HealthCheck, settings

    settings.register_profile(
        "ci",
        derandomize=True,
        deadline=None,
        suppress_health_check=[HealthCheck.too_slow],
    )
    settings.load_profile("ci")
except ImportError:
    pass
