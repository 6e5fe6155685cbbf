import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an AMD GPU (run via gpurun)")


def _have_gpu() -> bool:
    try:
        from starrocks_amd.engine import Engine
        return Engine.device_count() > 0
    except Exception:
        return False


@pytest.fixture(scope="session")
def engine():
    from starrocks_amd.engine import Engine
    if not _have_gpu():
        pytest.skip("no AMD GPU visible")
    eng = Engine(0)
    yield eng
    eng.close()
