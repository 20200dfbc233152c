import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu on a GPU box)"
    )


@pytest.fixture(scope="session")
def hip_backend():
    """Session-scoped HIP backend; skips cleanly when no GPU is present."""
    import torch

    if not torch.cuda.is_available():
        pytest.skip("no GPU available")
    from amgcl_amd.backend import make_backend

    return make_backend("hip")
