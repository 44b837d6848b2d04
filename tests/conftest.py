import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run via gpurun)")
    # safety net for the driver's -x -q runs: no single test may hang
    # the round's build/test tier (pytest-timeout is in the image)
    if config.getoption("--timeout", None) in (None, 0):
        config.option.timeout = 600


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def small_graph():
    """Deterministic small test graph: two dense-ish clusters + bridge."""
    from bigclam.io import planted_partition

    g, labels = planted_partition(3, 12, p_in=0.5, p_out=0.02, seed=7)
    return g


@pytest.fixture(scope="session")
def tiny_graph():
    from bigclam.io import build_graph

    edges = np.array(
        [[0, 1], [0, 2], [1, 2], [2, 3], [3, 4], [3, 5], [4, 5], [5, 6], [1, 0]]
    )
    return build_graph(edges)
