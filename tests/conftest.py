import subprocess
import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that must run on the GPU box (full-stack "
        "integration tier; this project has no GPU compute)")


@pytest.fixture(scope="session", autouse=True)
def built_native():
    """Build the native tree once per test session if it is stale."""
    subprocess.run(["make", "-j8"], cwd=REPO_ROOT, check=True,
                   capture_output=True)
    yield
