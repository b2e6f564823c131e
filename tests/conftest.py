import sys
from pathlib import Path

import pytest

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests requiring an MI355X GPU (run via gpurun)")


@pytest.fixture(autouse=True)
def _seed_everything():
    from fl4health_amd.utils.random import set_all_random_seeds

    set_all_random_seeds(42)
    yield
