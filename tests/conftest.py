import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need an MI355X (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def model_dir(tmp_path):
    return str(tmp_path / "model")


@pytest.fixture
def synthetic_classification():
    """Deterministic 4-class tabular problem + repeating input_fn."""
    torch.manual_seed(0)
    N, D, C = 512, 16, 4
    X = torch.randn(N, D)
    W = torch.randn(D, C)
    Y = (X @ W).argmax(dim=1)

    def input_fn():
        def gen():
            g = torch.Generator().manual_seed(7)
            while True:
                idx = torch.randint(0, N, (64,), generator=g)
                yield X[idx], Y[idx]

        return gen()

    return X, Y, input_fn
