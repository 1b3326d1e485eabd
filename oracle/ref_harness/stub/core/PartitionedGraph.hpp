/* TEST INFRASTRUCTURE ONLY — stub "core/PartitionedGraph.hpp" for oracle/_ref
 * (see stub/core/graph.hpp for the full story).  Supplies the members the
 * reference's op sources read: graph_, graph_chunks, has_mirror_at
 * (single-partition: every vertex is local, so always true). */
#ifndef NTS_REF_STUB_PARTITIONEDGRAPH_HPP
#define NTS_REF_STUB_PARTITIONEDGRAPH_HPP

#include <vector>

#include "core/graph.hpp"

struct PartitionedGraph {
  Graph<Empty> *graph_ = nullptr;
  std::vector<CSC_segment_pinned *> graph_chunks;
  VertexId owned_vertices = 0;
  /* dist-GAT members (core/PartitionedGraph.hpp surface): whole-graph CSC
   * over the owned dst range + the compressed mirror index
   * (generateMirrorIndex numbering, :295-305) */
  VertexId owned_edges = 0, owned_mirrors = 0;
  VertexId *column_offset = nullptr; /* [owned_vertices+1] local */
  VertexId *row_indices = nullptr;   /* [owned_edges] global src ids */
  VertexId *MirrorIndex = nullptr;   /* [global_vertices+1] */
  bool has_mirror_at(int partition, VertexId vid) {
    (void)partition;
    (void)vid;
    return true;
  }
  /* per-owned-master driver (core/PartitionedGraph.hpp:421-443; OpenMP in
   * the reference, serial here for determinism) */
  template <typename FN> void DistSchedulingMaster(FN fn) {
    for (VertexId d = graph_->gnnctx->p_v_s; d < graph_->gnnctx->p_v_e; ++d)
      fn(d, this);
  }
};

#endif
