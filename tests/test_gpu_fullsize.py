"""Full-size property tests (BASELINE config #2 scale): the oracle cannot run
the 114M-edge workload in test time, so parity at full size is pinned through
size-independent properties of the aggregation (the SURVEY §8c plan):

  - closed form on all-ones features: Y[d] = sum of incident norm-degree
    weights (checked against an independent fp64 np.add.at accumulation);
  - linearity: agg(a·x + b·y) == a·agg(x) + b·agg(y);
  - forward/backward adjointness: <agg_fwd(x), g> == <x, agg_bwd(g)>
    (CSC forward and CSR backward are transposes of the same weighted
    operator).
"""
import numpy as np
import pytest
import torch

from neutronstarlite_amd import graph as G

pytestmark = pytest.mark.gpu

V, E, F = 232_965, 114_000_000, 602


@pytest.fixture(scope="module")
def full():
    assert torch.cuda.is_available()
    from neutronstarlite_amd.ops import DeviceChunk, HipEngine
    dev = torch.device("cuda:0")
    edges = G.rmat_edges(V, E, seed=7)
    outd, ind = G.degrees(edges, V)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    ch = G.build_chunks(edges, w, np.array([0, V], dtype=np.uint32), 0)[0]
    # independent fp64 per-destination weight sums for the closed form
    colsum = np.zeros(V)
    np.add.at(colsum, edges[:, 1], w.astype(np.float64))
    eng = HipEngine()
    return {"dev": dev, "dch": DeviceChunk(ch, dev), "eng": eng,
            "colsum": colsum}


def _fwd(full, x):
    y = torch.zeros(x.shape[0], x.shape[1], device=full["dev"])
    full["eng"].csc_forward(full["dch"], x, y)
    torch.cuda.synchronize()
    return y


def test_all_ones_closed_form(full):
    f = 8  # width-independent property; small width keeps the check cheap
    x = torch.ones(V, f, device=full["dev"])
    y = _fwd(full, x)
    ref = torch.from_numpy(full["colsum"].astype(np.float32)).to(full["dev"])
    err = (y - ref[:, None]).abs()
    tol = 1e-4 * ref.abs()[:, None] + 1e-4  # fp32 sums over up-to-1M terms
    bad = int((err > tol).sum().item())
    assert bad == 0, f"{bad} entries off closed form; worst {err.max().item():.3e}"


def test_linearity_full_width(full):
    f = F
    gen = torch.Generator(device="cpu").manual_seed(42)
    x = torch.rand(V, f, generator=gen).to(full["dev"]) * 2 - 1
    y = torch.rand(V, f, generator=gen).to(full["dev"]) * 2 - 1
    a, b = 2.0, -3.0
    lhs = _fwd(full, (a * x + b * y).contiguous())
    rhs = a * _fwd(full, x) + b * _fwd(full, y)
    err = (lhs - rhs).abs()
    tol = 1e-4 * rhs.abs() + 1e-3
    bad = int((err > tol).sum().item())
    assert bad == 0, f"{bad} entries violate linearity; worst {err.max().item():.3e}"


def test_forward_backward_adjoint(full):
    """<A x, g> == <x, A^T g> ties the CSR backward to the CSC forward at
    full size (fp64 dot of fp32 results)."""
    f = 64
    gen = torch.Generator(device="cpu").manual_seed(1)
    x = torch.rand(V, f, generator=gen).to(full["dev"]) * 2 - 1
    g = torch.rand(V, f, generator=gen).to(full["dev"]) * 2 - 1
    ax = _fwd(full, x)
    atg = torch.zeros(V, f, device=full["dev"])
    full["eng"].csr_backward(full["dch"], g, atg)
    torch.cuda.synchronize()
    lhs = torch.dot(ax.double().flatten(), g.double().flatten()).item()
    rhs = torch.dot(x.double().flatten(), atg.double().flatten()).item()
    scale = abs(lhs) + abs(rhs) + 1.0
    assert abs(lhs - rhs) / scale < 1e-4, (lhs, rhs)
