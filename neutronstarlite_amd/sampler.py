"""Mini-batch neighbor sampling: the reference's reservoir
fan-out sampler + compacted sampled subgraph, rebuilt host-side and feeding
the SAME gfx950 gather kernels.

Reference semantics restated (citations into /root/reference):
  - per-layer fan-out sampling over the whole-graph CSC: a destination keeps
    min(deg, fanout) of its in-neighbors, uniformly chosen (reservoir
    algorithm, Sampler::reservoir_sample, core/ntsSampler.hpp:113-166);
  - layer i>0's destinations are layer i-1's compacted sources
    (sample_load_destination(i));
  - source compaction: distinct sampled sources get local ids in first-
    occurrence order; row_indices are rewritten to local ids
    (sampCSC::postprocessing, core/coocsc.hpp:62-89);
  - aggregation weights stay the FULL-graph norm degrees
    (MiniBatchFuseOp, core/ntsMiniBatchGraphOp.hpp:61-131:
    nts_norm_degree(graph, global_src, global_dst)).

The sampler here is vectorized numpy (per-edge random keys + segmented
top-fanout) rather than the reference's sequential reservoir loop: the
contract — <= fanout uniformly-chosen in-neighbors per destination — is the
same; the RNG sequence is not part of the parity surface (aggregation on a
GIVEN sampled subgraph is, and is tested against the oracle).
"""
from dataclasses import dataclass
from typing import List

import numpy as np


@dataclass
class SampledLayer:
    """One layer's compacted subgraph (sampCSC equivalent)."""
    dst: np.ndarray            # u32 [n_dst] global dst ids
    src: np.ndarray            # u32 [n_src] global src ids (first-occurrence)
    column_offset: np.ndarray  # u32 [n_dst+1]
    row_indices_local: np.ndarray  # u32 [E'] local src ids
    row_indices_global: np.ndarray  # u32 [E'] global src ids
    edge_weight: np.ndarray    # f32 [E'] full-graph norm-degree weights
    # backward CSR over local src ids (stable-by-src permutation of CSC)
    row_offset: np.ndarray         # u32 [n_src+1]
    column_indices_local: np.ndarray  # u32 [E'] local dst ids
    edge_weight_backward: np.ndarray  # f32 [E']

    @property
    def n_dst(self):
        return len(self.dst)

    @property
    def n_src(self):
        return len(self.src)

    @property
    def e_size(self):
        return len(self.row_indices_local)


def sample_layer(column_offset: np.ndarray, row_indices: np.ndarray,
                 dst: np.ndarray, fanout: int, outd: np.ndarray,
                 ind: np.ndarray, rng: np.random.Generator) -> SampledLayer:
    """Sample <= fanout in-neighbors for each dst over the whole-graph CSC,
    then compact sources and build the local CSC+CSR."""
    deg = (column_offset[dst + 1] - column_offset[dst]).astype(np.int64)
    # gather each dst's full neighbor index range
    n_edges_full = int(deg.sum())
    dst_of_edge = np.repeat(np.arange(len(dst), dtype=np.int64), deg)
    starts = column_offset[dst].astype(np.int64)
    seq = np.arange(n_edges_full, dtype=np.int64) - np.repeat(
        np.concatenate([[0], np.cumsum(deg)[:-1]]), deg)
    eidx = starts[dst_of_edge] + seq          # position in the full CSC
    keep_all = deg[dst_of_edge] <= fanout
    # uniform without replacement via random keys: for each dst take the
    # fanout smallest keys among its edges
    keys = rng.random(n_edges_full)
    keys[keep_all] = -1.0  # always kept
    order = np.lexsort((keys, dst_of_edge))
    rank_in_dst = np.arange(n_edges_full) - np.repeat(
        np.concatenate([[0], np.cumsum(deg)[:-1]]), deg)
    sel_sorted = rank_in_dst < fanout
    sel = np.zeros(n_edges_full, dtype=bool)
    sel[order] = sel_sorted
    eidx_s = eidx[sel]
    d_local = dst_of_edge[sel].astype(np.uint32)
    src_global = row_indices[eidx_s]

    # compacted sources, first-occurrence order (coocsc.hpp:62-89)
    uniq, first_pos = np.unique(src_global, return_index=True)
    order_first = np.argsort(first_pos, kind="stable")
    source = uniq[order_first].astype(np.uint32)
    remap = np.empty(len(uniq), dtype=np.uint32)
    remap[order_first] = np.arange(len(uniq), dtype=np.uint32)
    src_local = remap[np.searchsorted(uniq, src_global)]

    # local CSC
    n_dst = len(dst)
    counts = np.bincount(d_local, minlength=n_dst)
    col_off = np.zeros(n_dst + 1, dtype=np.uint32)
    col_off[1:] = np.cumsum(counts).astype(np.uint32)
    csc_order = np.argsort(d_local, kind="stable")
    ril = src_local[csc_order]
    rig = src_global[csc_order]
    dst_glob_of_edge = dst[d_local[csc_order]]
    w = (1.0 / (np.sqrt(outd[rig].astype(np.float32)) *
                np.sqrt(ind[dst_glob_of_edge].astype(np.float32)))
         ).astype(np.float32)

    # local CSR = stable-by-src permutation of the CSC order
    perm = np.argsort(ril, kind="stable")
    n_src = len(source)
    row_off = np.zeros(n_src + 1, dtype=np.uint32)
    row_off[1:] = np.cumsum(np.bincount(ril[perm], minlength=n_src)).astype(np.uint32)
    dst_local_of_csc = d_local[csc_order].astype(np.uint32)

    return SampledLayer(
        dst=dst.astype(np.uint32), src=source,
        column_offset=col_off, row_indices_local=ril.astype(np.uint32),
        row_indices_global=rig.astype(np.uint32), edge_weight=w,
        row_offset=row_off,
        column_indices_local=dst_local_of_csc[perm],
        edge_weight_backward=w[perm])


def sample_subgraph(column_offset: np.ndarray, row_indices: np.ndarray,
                    targets: np.ndarray, fanouts: List[int],
                    outd: np.ndarray, ind: np.ndarray,
                    seed: int = 0) -> List[SampledLayer]:
    """Layer-wise sampling: layer 0 destinations are the batch targets;
    layer i destinations are layer i-1's compacted sources
    (reservoir_sample's layer loop, ntsSampler.hpp:113-166).  Returned in
    sample order (layer 0 = output layer, like sampled_sgs)."""
    rng = np.random.default_rng(seed)
    layers = []
    dst = np.asarray(targets, dtype=np.uint32)
    for f in fanouts:
        layer = sample_layer(column_offset, row_indices, dst, int(f),
                             outd, ind, rng)
        layers.append(layer)
        dst = layer.src
    return layers
