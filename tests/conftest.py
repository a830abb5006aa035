import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU")


@pytest.fixture(scope="session", autouse=True)
def built_library():
    """Make sure lib_migbm.so exists before any test runs."""
    lib = REPO / "lightgbm_amd" / "lib" / "lib_migbm.so"
    if not lib.exists():
        subprocess.run(["make", "-j8"], cwd=REPO, check=True)
    return lib
