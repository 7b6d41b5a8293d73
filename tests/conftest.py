import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)"
    )


@pytest.fixture(autouse=True)
def _quiet_logs():
    import faabric_amd as fa

    fa.set_log_level(os.environ.get("LOG_LEVEL", "error"))
    yield
