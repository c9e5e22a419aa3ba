import os
import sys

import pytest
import torch

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X (or any ROCm) GPU")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def small_graph():
    """Deterministic synthetic power-law-ish CSR graph."""
    import numpy as np
    rng = np.random.default_rng(0)
    n = 500
    degs = np.minimum((rng.pareto(1.5, n) * 4).astype(int) + 1, 64)
    indptr = np.zeros(n + 1, dtype=np.int64)
    indptr[1:] = np.cumsum(degs)
    indices = rng.integers(0, n, indptr[-1], dtype=np.int64)
    return (torch.from_numpy(indptr), torch.from_numpy(indices))
