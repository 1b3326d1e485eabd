/* TEST INFRASTRUCTURE ONLY — stub "ntsSampler.hpp" for oracle/_ref
 * (see stub/core/graph.hpp).  Supplies the sampled-subgraph surface that
 * core/ntsMiniBatchGraphOp.hpp's loop bodies read — sampCSC's
 * dst()/src()/r_i() accessors and SampledSubgraph::compute_one_layer's
 * per-destination driver (the reference's is OpenMP over the same
 * iteration space; serial here for determinism).  The op's ARITHMETIC
 * (nts_comp/nts_acc with nts_norm_degree over the sampled indices,
 * ntsMiniBatchGraphOp.hpp:71-129) is compiled from the reference tree. */
#ifndef NTS_REF_STUB_SAMPLER_HPP
#define NTS_REF_STUB_SAMPLER_HPP

#include <vector>

#include "core/graph.hpp"

struct sampCSC {
  std::vector<VertexId> v_dst, v_src, c_o, r_indices;
  std::vector<VertexId> &dst() { return v_dst; }
  std::vector<VertexId> &src() { return v_src; }
  std::vector<VertexId> &c_o_ref() { return c_o; }
  VertexId r_i(VertexId off) { return r_indices[off]; }
};

struct SampledSubgraph {
  std::vector<sampCSC *> sampled_sgs;
  template <typename FN> void compute_one_layer(FN fn, int layer) {
    sampCSC *sg = sampled_sgs[layer];
    for (VertexId d = 0; d < (VertexId)sg->v_dst.size(); ++d)
      fn(d, sg->c_o, sg->r_indices);
  }
};

#endif
