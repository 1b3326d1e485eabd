"""CPU-side invariants of the GAT layer plumbing: the CSC->CSR edge
permutation used to carry attention weights, and an oracle composition of
the full GAT layer forward (reference chain, GAT_GPU_DIST.hpp:191-215)."""
import numpy as np

import oracle
from neutronstarlite_amd import graph as G


def _setup(v=400, e=6000, seed=13):
    edges = G.rmat_edges(v, e, seed=seed)
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    ch = G.build_chunks(edges, w, np.array([0, v], dtype=np.uint32), 0)[0]
    return v, ch


def test_csr_is_stable_src_perm_of_csc():
    v, ch = _setup()
    src_of_csc = ch.row_indices.astype(np.int64)
    perm = np.argsort(src_of_csc, kind="stable")
    deg = np.diff(ch.column_offset.astype(np.int64))
    dst_of_csc = np.repeat(np.arange(v, dtype=np.int64), deg)
    # applying the perm to CSC edges must reproduce the CSR arrays exactly
    assert np.array_equal(ch.column_indices.astype(np.int64),
                          dst_of_csc[perm])
    assert np.array_equal(ch.edge_weight_backward,
                          ch.edge_weight_forward[perm])
    # and the CSR row structure matches
    counts = np.bincount(src_of_csc[perm], minlength=v)
    assert np.array_equal(np.diff(ch.row_offset.astype(np.int64)), counts)


def gat_forward_oracle(ch, v, h, a_src, a_dst, slope=0.2):
    """Full GAT layer forward composed from oracle pieces (fp32)."""
    f = h.shape[1]
    E = ch.edge_size
    s_src = (h @ a_src).astype(np.float32).reshape(v, 1)
    s_dst = (h @ a_dst).astype(np.float32).reshape(v, 1)
    mi = np.arange(v, dtype=np.uint32)
    m_src = np.zeros((E, 1), np.float32)
    oracle.scatter_src_to_msg(m_src, s_src, ch.row_indices, ch.column_offset,
                              mi, v, 1)
    m_dst = np.zeros((E, 1), np.float32)
    oracle.scatter_dst_to_msg(m_dst, s_dst, ch.column_offset, v, 1)
    e_val = m_src + m_dst
    e_val = np.where(e_val > 0, e_val, slope * e_val).astype(np.float32)
    s = np.zeros((E, 1), np.float32)
    cached = np.zeros((E, 1), np.float32)
    oracle.edge_softmax_forward(s, e_val, cached, ch.column_offset, v, 1)
    y = oracle.csc_forward(ch.column_offset, ch.row_indices,
                           np.ascontiguousarray(s[:, 0]), h, 0, v, f)
    return y, s


def test_gat_oracle_composition_sane():
    v, ch = _setup()
    f = 16
    rng = np.random.default_rng(3)
    h = rng.uniform(-1, 1, size=(v, f)).astype(np.float32)
    a_src = rng.uniform(-1, 1, size=f).astype(np.float32)
    a_dst = rng.uniform(-1, 1, size=f).astype(np.float32)
    y, s = gat_forward_oracle(ch, v, h, a_src, a_dst)
    # attention-convexity: each output row lies in the convex hull of its
    # neighbors' rows => bounded by per-column min/max over sources
    deg = np.diff(ch.column_offset.astype(np.int64))
    d = int(np.argmax(deg))
    cols = ch.row_indices[ch.column_offset[d]:ch.column_offset[d + 1]]
    assert np.all(y[d] <= h[cols].max(0) + 1e-5)
    assert np.all(y[d] >= h[cols].min(0) - 1e-5)
