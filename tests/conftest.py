import os
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO not in sys.path:
    sys.path.insert(0, REPO)

REFERENCE = "/root/reference"  # only read at TEST time in this container,
# never at run time on the GPU box; gpu-marked tests use committed fixtures.


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a real MI355X (run via gpurun / driver)")


def pytest_collection_modifyitems(config, items):
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def small_graph():
    """Seeded power-law graph small enough for exhaustive oracle checks."""
    from neutronstarlite_amd import graph as G
    v, e = 2000, 16000
    edges = G.rmat_edges(v, e, seed=7)
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    return {"v": v, "edges": edges, "outd": outd, "ind": ind, "w": w}


@pytest.fixture(scope="session")
def cora():
    """The vendored Cora edge list (data/cora.2708.edge.self: 13 566 edges,
    Gemini binary), loaded from the committed fixture copy."""
    path = os.path.join(REPO, "tests", "golden", "cora.2708.edge.self.npy")
    edges = np.load(path)
    return {"v": 2708, "edges": edges}
