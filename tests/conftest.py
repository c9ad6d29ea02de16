import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X (run via gpurun)")


@pytest.fixture(scope="session", autouse=True)
def built_libs():
    """Build the oracle (and the HIP lib if sources are newer) once per run.

    On a GPU box the .so files arrive prebuilt with the snapshot; make is a
    no-op then.
    """
    subprocess.run(["make", "-s", "-C", os.path.join(REPO, "oracle")], check=True)
    csrc = os.path.join(REPO, "m3_amd", "csrc")
    if os.path.exists(os.path.join(csrc, "Makefile")):
        subprocess.run(["make", "-s", "-C", csrc], check=True)
