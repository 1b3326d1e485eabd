"""Host-side graph plumbing: Gemini IO, RMAT, partitioning, chunks, items."""
import numpy as np
import pytest

from neutronstarlite_amd import graph as G


def test_gemini_io_roundtrip(tmp_path):
    edges = G.rmat_edges(100, 500, seed=3)
    p = tmp_path / "g.edge"
    edges.astype(np.uint32).tofile(p)
    back = G.load_gemini_edges(str(p))
    assert np.array_equal(edges, back)


def test_rmat_deterministic_and_skewed():
    e1 = G.rmat_edges(1 << 12, 50000, seed=7)
    e2 = G.rmat_edges(1 << 12, 50000, seed=7)
    assert np.array_equal(e1, e2)
    outd = np.bincount(e1[:, 0], minlength=1 << 12)
    # power-law: top 1% of vertices own a large share of edges
    top = np.sort(outd)[-41:].sum()
    assert top > 0.05 * len(e1)
    # self loops appended
    assert np.array_equal(e1[-(1 << 12):, 0], e1[-(1 << 12):, 1])


def test_partition_offsets_balance():
    edges = G.rmat_edges(4096, 60000, seed=7)
    offs = G.partition_offsets(edges, 4096, 4)
    assert offs[0] == 0 and offs[-1] == 4096
    outd = np.bincount(edges[:, 0], minlength=4096)
    loads = [outd[offs[i]:offs[i + 1]].sum() for i in range(4)]
    assert max(loads) < 2.0 * (sum(loads) / 4)


def test_partition_offsets_hub_dominated():
    """One vertex carrying almost all load must yield a valid (if
    unbalanced) partition, not an assertion: duplicate
    searchsorted boundaries are nudged forward."""
    v, parts = 64, 4
    src = np.concatenate([np.zeros(10000, dtype=np.uint32),
                          np.arange(v, dtype=np.uint32)])
    dst = np.concatenate([np.arange(10000, dtype=np.uint32) % v,
                          np.arange(v, dtype=np.uint32)])
    edges = np.stack([src, dst], axis=1)
    offs = G.partition_offsets(edges, v, parts)
    d = np.diff(offs.astype(np.int64))
    assert offs[0] == 0 and offs[-1] == v and (d > 0).all()


def test_chunks_cover_all_edges():
    v, parts = 1024, 4
    edges = G.rmat_edges(v, 20000, seed=7)
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    offs = G.partition_offsets(edges, v, parts)
    total = 0
    for r in range(parts):
        chunks = G.build_chunks(edges, w, offs, r)
        assert len(chunks) == parts
        for k, ch in enumerate(chunks):
            total += ch.edge_size
            assert ch.column_offset[-1] == ch.edge_size
            assert ch.row_offset[-1] == ch.edge_size
            if ch.edge_size:
                assert ch.row_indices.min() >= ch.src_s
                assert ch.row_indices.max() < ch.src_e
                assert ch.column_indices.min() >= ch.dst_s
                assert ch.column_indices.max() < ch.dst_e
            # CSC and CSR hold the same multiset of weights
            assert np.isclose(ch.edge_weight_forward.sum(),
                              ch.edge_weight_backward.sum(), rtol=1e-5)
    assert total == len(edges)


def test_single_partition_chunk_is_whole_graph():
    v = 300
    edges = G.rmat_edges(v, 2000, seed=7)
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    ch = G.build_chunks(edges, w, np.array([0, v], dtype=np.uint32), 0)[0]
    assert ch.edge_size == len(edges)
    # per-dst in-degree matches CSC column lengths (the reference's own
    # structural check, test/testcsr.cpp:40-45)
    ind_raw = np.bincount(edges[:, 1], minlength=v)
    assert np.array_equal(np.diff(ch.column_offset.astype(np.int64)), ind_raw)


@pytest.mark.parametrize("split", [1, 8, 512])
def test_work_items_cover_and_flag(split):
    v = 200
    edges = G.rmat_edges(v, 5000, seed=7)
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    ch = G.build_chunks(edges, w, np.array([0, v], dtype=np.uint32), 0)[0]
    vtx, st, cnt = G.build_work_items(ch.column_offset, split=split)
    deg = np.diff(ch.column_offset.astype(np.int64))
    assert cnt.sum() == deg.sum()
    assert cnt.max() <= split
    vids = vtx & 0x7FFFFFFF
    flags = vtx >> 31
    # every edge covered exactly once, in the right column
    for i in range(len(vtx)):
        lo, hi = ch.column_offset[vids[i]], ch.column_offset[vids[i] + 1]
        assert lo <= st[i] and st[i] + cnt[i] <= hi
    # flags: set iff vertex has >1 item
    n_items = np.bincount(vids, minlength=v)
    assert np.all((n_items[vids] > 1) == (flags == 1))


def test_degenerate_sizes():
    """Edge cases the reference never guards: single-vertex graphs, an
    all-self-loop graph, and a vertex range with zero edges must flow
    through degrees/weights/chunks without error."""
    # single vertex, single self loop
    edges = np.array([[0, 0]], dtype=np.uint32)
    outd, ind = G.degrees(edges, 1)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    ch = G.build_chunks(edges, w, np.array([0, 1], dtype=np.uint32), 0)[0]
    assert ch.edge_size == 1 and ch.column_offset[-1] == 1
    # all-self-loop graph
    v = 17
    edges = np.stack([np.arange(v, dtype=np.uint32)] * 2, axis=1)
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    assert np.allclose(w, 1.0)
    ch = G.build_chunks(edges, w, np.array([0, v], dtype=np.uint32), 0)[0]
    import oracle
    x = np.random.default_rng(0).normal(size=(v, 3)).astype(np.float32)
    y = oracle.csc_forward(ch.column_offset, ch.row_indices,
                           ch.edge_weight_forward, x, 0, v, 3)
    assert np.allclose(y, x, rtol=1e-6)
    # zero-edge partition range: chunks exist with empty CSC
    edges = np.array([[0, 1]], dtype=np.uint32)
    outd, ind = G.degrees(edges, 8)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    offs = np.array([0, 4, 8], dtype=np.uint32)
    chunks = G.build_chunks(edges, w, offs, 1)  # rank 1 owns dst 4..8: none
    assert all(ch.edge_size == 0 for ch in chunks)
    assert all(ch.column_offset[-1] == 0 for ch in chunks)
