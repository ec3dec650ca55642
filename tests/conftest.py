import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires a real MI355X GPU")


@pytest.fixture(scope="session", autouse=True)
def _build_oracle():
    """Ensure the oracle .so exists (cheap no-op rebuild if current)."""
    subprocess.run(["make", "-s", "-C", os.path.join(REPO, "oracle")], check=True)
