import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
if str(REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X GPU (run via gpurun)"
    )


@pytest.fixture
def repo_root() -> Path:
    return REPO_ROOT


@pytest.fixture(scope="session", autouse=True)
def build_native():
    """Every suite needs the native tree once per session (no-op when
    up to date)."""
    import subprocess
    subprocess.run(["make", "-C", str(REPO_ROOT / "native"), "-j8"],
                   check=True, capture_output=True)
