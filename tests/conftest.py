import pytest

from rl_replicas_amd.utils import set_seed_for_libraries


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run with -m gpu)")
    config.addinivalue_line("markers", "slow: takes tens of seconds on CPU")


def pytest_collection_modifyitems(config, items):
    """Skip gpu-marked tests automatically when no GPU is present."""
    import torch

    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def seed() -> int:
    return 0


@pytest.fixture(autouse=True)
def set_seed(seed: int):
    """Determinism is the foundation of the test strategy (reference
    tests/conftest.py:6-17): every test runs under a fixed seed with
    torch deterministic algorithms enabled."""
    set_seed_for_libraries(seed)
