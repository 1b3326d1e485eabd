"""GPU parity of the mini-batch op (MiniBatchFuseOp equivalent) against the
CPU oracle on sampled subgraphs, including a 2-layer chained pass."""
import numpy as np
import pytest
import torch

import oracle
from neutronstarlite_amd import graph as G
from neutronstarlite_amd.sampler import sample_subgraph

pytestmark = pytest.mark.gpu


def assert_close(got, ref, name=""):
    got = got.cpu().numpy()
    err = np.abs(got - ref)
    bad = err > 1e-4 * np.abs(ref) + 1e-5
    assert not bad.any(), f"{name}: {bad.sum()}/{bad.size} out of tol"


def test_minibatch_fwd_bwd_two_layers():
    from neutronstarlite_amd.ops import HipEngine, MiniBatchFuseOp
    dev = torch.device("cuda:0")
    v, e, f = 3000, 60000, 40
    edges = G.rmat_edges(v, e, seed=7)
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    ch = G.build_chunks(edges, w, np.array([0, v], dtype=np.uint32), 0)[0]
    rng = np.random.default_rng(8)
    targets = rng.choice(v, size=256, replace=False).astype(np.uint32)
    layers = sample_subgraph(ch.column_offset, ch.row_indices, targets,
                             [10, 5], outd, ind, seed=9)

    eng = HipEngine()
    # innermost layer first: features of layer-1's compacted sources
    x1 = rng.uniform(-1, 1, size=(layers[1].n_src, f)).astype(np.float32)
    op1 = MiniBatchFuseOp(layers[1], dev, eng)
    h1 = op1.forward(torch.from_numpy(x1).to(dev))   # rows = layers[1].dst
    torch.cuda.synchronize()
    h1_ref = oracle.csc_forward(layers[1].column_offset,
                                layers[1].row_indices_local,
                                layers[1].edge_weight, x1, 0,
                                layers[1].n_dst, f)
    assert_close(h1, h1_ref, "layer1 fwd")

    # layer 0 consumes layer 1's output (dst of layer1 == src of layer0)
    op0 = MiniBatchFuseOp(layers[0], dev, eng)
    y = op0.forward(h1.contiguous())
    torch.cuda.synchronize()
    y_ref = oracle.csc_forward(layers[0].column_offset,
                               layers[0].row_indices_local,
                               layers[0].edge_weight, h1_ref, 0,
                               layers[0].n_dst, f)
    assert_close(y, y_ref, "layer0 fwd")

    gy = rng.uniform(-1, 1, size=(layers[0].n_dst, f)).astype(np.float32)
    gx = op0.backward(torch.from_numpy(gy).to(dev))
    torch.cuda.synchronize()
    gx_ref = oracle.csr_backward(layers[0].row_offset,
                                 layers[0].column_indices_local,
                                 layers[0].edge_weight_backward, gy, 0,
                                 layers[0].n_src, f)
    assert_close(gx, gx_ref, "layer0 bwd")
