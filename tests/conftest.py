import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

# Import torch (its bundled HIP runtime) BEFORE any test dlopens the
# futuresdr_amd C-ABI library: loading /opt/rocm's libamdhip64 first makes
# torch's lazy HIP init report no devices (see bench.py header). Harmless
# on CPU-only boxes.
try:
    import torch  # noqa: F401
except ImportError:
    pass


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a real MI355X (run via gpurun / driver)")


@pytest.fixture(scope="session")
def oracle_lib():
    import oracle
    if not os.path.exists(os.path.join(REPO, "oracle", "liboracle.so")):
        subprocess.run(["make", "-s", "-C", os.path.join(REPO, "oracle")],
                       check=True)
    return oracle


@pytest.fixture(scope="session")
def fsdr():
    import futuresdr_amd
    so = os.path.join(REPO, "futuresdr_amd", "libfutursdr_hip.so")
    if not os.path.exists(so):
        futuresdr_amd.build()
    return futuresdr_amd


@pytest.fixture(scope="session")
def gpu(fsdr):
    if fsdr.device_count() < 1:
        pytest.fail("gpu-marked test ran without a HIP device")
    return fsdr
