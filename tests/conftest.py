import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a visible MI355X GPU")
    # build native libs if missing (no-op when current)
    if not os.path.exists(os.path.join(REPO, "citus_amd", "libcstripe.so")):
        subprocess.check_call(["make", "-C", os.path.join(REPO, "citus_amd", "csrc")])
    if not os.path.exists(os.path.join(REPO, "oracle", "liboracle.so")):
        subprocess.check_call(["make", "-C", os.path.join(REPO, "oracle")])


@pytest.fixture(scope="session")
def golden_dir():
    return os.path.join(REPO, "tests", "golden")


@pytest.fixture(scope="session")
def expected(golden_dir):
    import json
    with open(os.path.join(golden_dir, "expected.json")) as f:
        return json.load(f)


def q6_preds(ca, exp):
    q6 = exp["q6"]
    return [(5, ca.PRED_GE, q6["shipdate_ge"]), (5, ca.PRED_LT, q6["shipdate_lt"]),
            (3, ca.PRED_GE, 5), (3, ca.PRED_LE, 7), (1, ca.PRED_LT, 2400)]

# torch and the system ROCm runtime disagree when libcstripe (ctypes ->
# system libamdhip64) initializes HIP first in a process: torch.cuda then
# enumerates 0 devices (torch bundles its own HIP runtime). Initializing
# torch's view FIRST makes both stacks coexist; harmless on CPU-only boxes.
try:
    import torch
    torch.cuda.is_available()
except Exception:
    pass
