import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO_ROOT)
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test needs a real MI355X GPU (run via gpurun)"
    )


def load_golden(name):
    import numpy as np

    path = os.path.join(os.path.dirname(__file__), "golden", f"{name}.npz")
    z = np.load(path)
    n = int(z["n_runs"][0])
    runs = [
        (z[f"run{i}_data"].tobytes(), z[f"run{i}_index"].tobytes())
        for i in range(n)
    ]
    keep = (z["keep_data"].tobytes(), z["keep_index"].tobytes())
    drop = (z["drop_data"].tobytes(), z["drop_index"].tobytes())
    return runs, keep, drop


GOLDEN_CASES = [
    "basic",
    "ts_shuffle",
    "tie_ts",
    "all_tombstones",
    "empty_run",
    "single_run",
    "ragged",
    "neg_ts",
    "sixteen_runs",
    "disjoint",
]


@pytest.fixture(params=GOLDEN_CASES)
def golden_case(request):
    return request.param, load_golden(request.param)
