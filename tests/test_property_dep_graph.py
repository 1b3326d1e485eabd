"""Property tests for the multi-partition GAT mirror structures
(dist_gat.build_dep_graph): the compressed mirror index, per-partition
slot slices and reindexed CSC/CSR must satisfy the generateMirrorIndex
invariants (PartitionedGraph.hpp:295-305) on arbitrary graphs."""
import numpy as np
from hypothesis import given, settings, strategies as st

from neutronstarlite_amd import graph as G
from neutronstarlite_amd.dist_gat import build_dep_graph


@settings(max_examples=25, deadline=None)
@given(st.integers(10, 200), st.integers(1, 800), st.integers(2, 4),
       st.integers(0, 2**31 - 1))
def test_dep_graph_invariants(v, e, parts, seed):
    rng = np.random.default_rng(seed)
    edges = np.stack([rng.integers(0, v, e), rng.integers(0, v, e)],
                     axis=1).astype(np.uint32)
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    if v < parts:
        return
    offs = G.partition_offsets(edges, v, parts)
    for rank in range(parts):
        dg = build_dep_graph(edges, w, offs, rank, v)
        lo, hi = int(offs[rank]), int(offs[rank + 1])
        owned = edges[(edges[:, 1] >= lo) & (edges[:, 1] < hi)]
        # mirrors are exactly the distinct sources of owned edges, ascending
        assert np.array_equal(dg.mirrors, np.unique(owned[:, 0]))
        # compressed index round-trips (generateMirrorIndex numbering)
        assert np.array_equal(dg.mirrors[dg.mirror_index[dg.mirrors]],
                              dg.mirrors)
        # partition slot slices tile the mirror range in order
        assert dg.part_slice[0][0] == 0
        assert dg.part_slice[-1][1] == dg.n_mirrors
        for k in range(parts - 1):
            assert dg.part_slice[k][1] == dg.part_slice[k + 1][0]
        for k in range(parts):
            a, b = dg.part_slice[k]
            m = dg.mirrors[a:b]
            assert ((m >= offs[k]) & (m < offs[k + 1])).all()
        # reindexed CSC covers every owned edge once; reindex is consistent
        assert dg.column_offset[-1] == len(owned)
        assert np.array_equal(dg.mirrors[dg.row_indices_m], dg.row_indices)
        # CSR of the reindexed graph is the stable-by-src-slot permutation
        assert np.array_equal(dg.row_indices_m[dg.csr_from_csc],
                              np.sort(dg.row_indices_m, kind="stable"))
        assert dg.row_offset_m[-1] == len(owned)
        # per-mirror-row counts match
        cnt = np.bincount(dg.row_indices_m.astype(np.int64),
                          minlength=dg.n_mirrors)
        assert np.array_equal(np.diff(dg.row_offset_m.astype(np.int64)), cnt)
