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


@pytest.mark.timeout(300)
def test_sampled_training_step_learns():
    """End-to-end sampled training (DESIGN §9.4 / bench --model
    gcn-sample-train): GPU-resident sampling + resident-feature gather +
    2-layer GCN through the minibatch autograd bridge + Adam; the loss on a
    learnable synthetic labeling must drop."""
    from neutronstarlite_amd.ops import (MiniBatchFuseOp, _u32_cuda,
                                         minibatch_aggregate)
    from neutronstarlite_amd.sampler_gpu import sample_subgraph_gpu

    dev = torch.device("cuda:0")
    v, e, f, f1, ncls = 4000, 60000, 32, 16, 5
    edges = G.rmat_edges(v, e, seed=9)
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    ch = G.build_chunks(edges, w, np.array([0, v], dtype=np.uint32), 0)[0]
    d_coff = _u32_cuda(ch.column_offset, dev)
    d_rows = _u32_cuda(ch.row_indices, dev)
    d_outd = torch.from_numpy(outd.astype(np.int64)).to(dev)
    d_ind = torch.from_numpy(ind.astype(np.int64)).to(dev)
    rng = np.random.default_rng(11)
    x = torch.from_numpy(rng.uniform(-1, 1, (v, f)).astype(np.float32)).to(dev)
    # learnable labels: a fixed random projection of the features
    proj = torch.from_numpy(rng.normal(size=(f, ncls)).astype(np.float32)).to(dev)
    labels = (x @ proj).argmax(1)
    W0 = (torch.rand(f, f1, device=dev) * 0.2 - 0.1).requires_grad_(True)
    W1 = (torch.rand(f1, ncls, device=dev) * 0.2 - 0.1).requires_grad_(True)
    opt = torch.optim.Adam([W0, W1], lr=5e-2)
    losses = []
    for i in range(30):
        targets = torch.from_numpy(
            rng.choice(v, 512, replace=False).astype(np.int32)).to(dev)
        from neutronstarlite_amd import shim
        layers = sample_subgraph_gpu(shim.Stream.wrap_torch_current(),
                                     d_coff, d_rows, targets, [10, 5],
                                     d_outd, d_ind, seed=100 + i)
        ops_ = [MiniBatchFuseOp(ly, dev) for ly in layers]
        x_s = x.index_select(0, layers[-1].src)
        opt.zero_grad(set_to_none=True)
        h = minibatch_aggregate(x_s, ops_[-1])
        h = torch.relu(h @ W0)
        h = minibatch_aggregate(h, ops_[0])
        out = torch.log_softmax(h @ W1, 1)
        loss = torch.nn.functional.nll_loss(
            out, labels[targets.to(torch.int64)])
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert np.mean(losses[-5:]) < np.mean(losses[:5]) - 0.05, losses[:3] + losses[-3:]
