import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO not in sys.path:
    sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X (run via gpurun)")


def _build_oracle():
    lib = os.path.join(REPO, "oracle", "liborc.so")
    if not os.path.exists(lib):
        import subprocess
        subprocess.run(["make", "-C", os.path.join(REPO, "oracle"), "liborc.so"],
                       check=True, capture_output=True)


_build_oracle()


@pytest.fixture
def tmp_job_dir(tmp_path):
    out = tmp_path / "out"
    out.mkdir()
    return tmp_path, out
