"""Property-based invariants (hypothesis) of the host graph plumbing:
random edge lists -> chunk construction, work items and weights must hold
their contracts for ANY input, not just the seeded fixtures."""
import numpy as np
from hypothesis import given, settings, strategies as st

import oracle
from neutronstarlite_amd import graph as G


@st.composite
def edge_lists(draw):
    v = draw(st.integers(min_value=2, max_value=120))
    n = draw(st.integers(min_value=0, max_value=400))
    src = draw(st.lists(st.integers(0, v - 1), min_size=n, max_size=n))
    dst = draw(st.lists(st.integers(0, v - 1), min_size=n, max_size=n))
    edges = np.array(list(zip(src, dst)), dtype=np.uint32).reshape(-1, 2)
    return v, edges


@given(edge_lists(), st.integers(1, 4))
@settings(max_examples=40, deadline=None)
def test_chunks_partition_edges_exactly(data, parts):
    v, edges = data
    if len(edges) == 0:
        return
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    parts = min(parts, v)
    try:
        offs = G.partition_offsets(edges, v, parts)
    except AssertionError:
        return  # empty partition for this draw — partitioner refuses, fine
    total = 0
    for r in range(parts):
        for ch in G.build_chunks(edges, w, offs, r):
            total += ch.edge_size
            assert ch.column_offset[-1] == ch.edge_size
            assert ch.row_offset[-1] == ch.edge_size
            if ch.edge_size:
                assert ch.row_indices.min() >= ch.src_s
                assert ch.row_indices.max() < ch.src_e
            # CSR is the stable-by-src permutation of the CSC
            perm = np.argsort(ch.row_indices.astype(np.int64), kind="stable")
            assert np.array_equal(ch.edge_weight_backward,
                                  ch.edge_weight_forward[perm])
    assert total == len(edges)


@given(edge_lists(), st.integers(1, 64))
@settings(max_examples=40, deadline=None)
def test_work_items_partition_columns(data, split):
    v, edges = data
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    ch = G.build_chunks(edges, w, np.array([0, v], dtype=np.uint32), 0)[0]
    vtx, starts, cnt = G.build_work_items(ch.column_offset, split=split)
    assert cnt.sum() == ch.edge_size
    # items of one vertex tile its column contiguously without overlap
    vids = vtx & 0x7FFFFFFF
    for u in np.unique(vids):
        mine = np.sort(starts[vids == u])
        lo, hi = ch.column_offset[u], ch.column_offset[u + 1]
        assert mine[0] == lo
        ends = mine + cnt[vids == u][np.argsort(starts[vids == u])]
        assert ends[-1] == hi
        assert np.all(mine[1:] == ends[:-1])


@given(edge_lists())
@settings(max_examples=30, deadline=None)
def test_oracle_forward_equals_dense_matmul(data):
    v, edges = data
    if len(edges) == 0:
        return
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    ch = G.build_chunks(edges, w, np.array([0, v], dtype=np.uint32), 0)[0]
    f = 3
    rng = np.random.default_rng(0)
    x = rng.uniform(-1, 1, size=(v, f)).astype(np.float32)
    y = oracle.csc_forward(ch.column_offset, ch.row_indices,
                           ch.edge_weight_forward, x, 0, v, f)
    A = np.zeros((v, v))
    np.add.at(A, (edges[:, 1], edges[:, 0]), w.astype(np.float64))
    assert np.allclose(y, A @ x, rtol=1e-4, atol=1e-5)


@given(edge_lists(), st.integers(1, 12))
@settings(max_examples=25, deadline=None)
def test_host_sampler_contract(data, fanout):
    """Host sampler: <= fanout slots per destination, all real edge slots,
    for arbitrary graphs (multigraphs included)."""
    from neutronstarlite_amd.sampler import sample_layer
    v, edges = data
    if len(edges) == 0:
        return
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    ch = G.build_chunks(edges, w, np.array([0, v], dtype=np.uint32), 0)[0]
    rng = np.random.default_rng(0)
    dst = np.unique(rng.integers(0, v, size=min(v, 16)).astype(np.uint32))
    ly = sample_layer(ch.column_offset, ch.row_indices, dst, fanout,
                      outd, ind, rng)
    deg_full = (ch.column_offset[dst + 1]
                - ch.column_offset[dst]).astype(np.int64)
    deg_s = np.diff(ly.column_offset.astype(np.int64))
    assert np.array_equal(deg_s, np.minimum(deg_full, fanout))
    assert np.array_equal(ly.src[ly.row_indices_local], ly.row_indices_global)
