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
  bool has_mirror_at(int partition, VertexId vid) {
    (void)partition;
    (void)vid;
    return true;
  }
};

#endif
