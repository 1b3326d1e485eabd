"""Pins the oracle's GAT edge-path functions against independent numpy
implementations (softmax identities, scatter/gather by explicit indexing)."""
import numpy as np
import pytest

import oracle
from neutronstarlite_amd import graph as G


@pytest.fixture(scope="module")
def g():
    v, e = 300, 4000
    edges = G.rmat_edges(v, e, seed=11)
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    ch = G.build_chunks(edges, w, np.array([0, v], dtype=np.uint32), 0)[0]
    # compressed mirror index: position of each global src in the sorted
    # unique src set (PartitionedGraph::generateMirrorIndex semantics,
    # PartitionedGraph.hpp:295-305)
    uniq = np.unique(ch.row_indices)
    mirror_index = np.zeros(v, dtype=np.uint32)
    mirror_index[uniq] = np.arange(len(uniq), dtype=np.uint32)
    return {"v": v, "ch": ch, "uniq": uniq, "mi": mirror_index}


def test_scatter_gather_src_mirror(g):
    ch, mi, uniq = g["ch"], g["mi"], g["uniq"]
    f, E = 5, ch.edge_size
    rng = np.random.default_rng(0)
    mirror = rng.normal(size=(len(uniq), f)).astype(np.float32)
    msg = np.zeros((E, f), dtype=np.float32)
    oracle.scatter_src_to_msg(msg, mirror, ch.row_indices, ch.column_offset,
                              mi, ch.dst_n, f)
    # independent: edge e in CSC order has source row_indices[e]
    assert np.array_equal(msg, mirror[mi[ch.row_indices]])
    back = np.zeros_like(mirror)
    oracle.gather_msg_to_src(back, msg, ch.row_indices, ch.column_offset, mi,
                             ch.dst_n, f)
    ref = np.zeros_like(mirror)
    np.add.at(ref, mi[ch.row_indices], msg)
    assert np.allclose(back, ref, rtol=1e-5, atol=1e-6)


def test_scatter_gather_dst(g):
    ch = g["ch"]
    f, E = 3, ch.edge_size
    rng = np.random.default_rng(1)
    dstf = rng.normal(size=(ch.dst_n, f)).astype(np.float32)
    msg = np.zeros((E, f), dtype=np.float32)
    oracle.scatter_dst_to_msg(msg, dstf, ch.column_offset, ch.dst_n, f)
    dst_of_edge = np.repeat(np.arange(ch.dst_n),
                            np.diff(ch.column_offset.astype(np.int64)))
    assert np.array_equal(msg, dstf[dst_of_edge])
    acc = np.zeros_like(dstf)
    oracle.gather_msg_to_dst(acc, msg, ch.column_offset, ch.dst_n, f)
    ref = np.zeros_like(dstf)
    np.add.at(ref, dst_of_edge, msg)
    assert np.allclose(acc, ref, rtol=1e-5, atol=1e-6)
    # scatter_grad_back accumulates dst grad onto each incident edge
    mg = np.zeros((E, f), dtype=np.float32)
    oracle.scatter_grad_back_to_msg(dstf, mg, ch.column_offset, ch.dst_n, f)
    assert np.array_equal(mg, dstf[dst_of_edge])


def test_edge_softmax_identities(g):
    ch = g["ch"]
    f, E = 1, ch.edge_size
    rng = np.random.default_rng(2)
    scores = rng.normal(scale=2.0, size=(E, f)).astype(np.float32)
    out = np.zeros_like(scores)
    cached = np.zeros_like(scores)
    oracle.edge_softmax_forward(out, scores, cached, ch.column_offset,
                                ch.dst_n, f)
    assert np.array_equal(out, cached)
    dst_of_edge = np.repeat(np.arange(ch.dst_n),
                            np.diff(ch.column_offset.astype(np.int64)))
    # per-dst sums to 1 where the dst has edges
    sums = np.zeros(ch.dst_n)
    np.add.at(sums, dst_of_edge, out[:, 0].astype(np.float64))
    deg = np.diff(ch.column_offset.astype(np.int64))
    assert np.allclose(sums[deg > 0], 1.0, atol=1e-5)
    # independent softmax (fp64, same no-max-subtraction form)
    ex = np.exp(scores[:, 0].astype(np.float64))
    den = np.zeros(ch.dst_n)
    np.add.at(den, dst_of_edge, ex)
    assert np.allclose(out[:, 0], ex / den[dst_of_edge], rtol=1e-4, atol=1e-6)

    # backward: gradient of softmax, checked against finite differences of
    # a scalar loss L = sum(c * s) for random c
    c = rng.normal(size=(E, f)).astype(np.float32)
    gin = np.zeros_like(scores)
    oracle.edge_softmax_backward(gin, c, cached, ch.column_offset, ch.dst_n, f)
    s = out[:, 0].astype(np.float64)
    dot = np.zeros(ch.dst_n)
    np.add.at(dot, dst_of_edge, c[:, 0] * s)
    ref = c[:, 0] * s - dot[dst_of_edge] * s
    assert np.allclose(gin[:, 0], ref, rtol=1e-4, atol=1e-6)
