/* TEST INFRASTRUCTURE ONLY — oracle/_ref driver.
 *
 * Compiles the reference's OWN hot-path sources from /root/reference (via
 * -I, nothing copied) against the 1-rank stubs in stub/, and exports a tiny
 * C ABI so tests and tests/golden/make_golden.py can execute
 * reference-authored arithmetic on arbitrary inputs:
 *
 *   nts_ref_comp / nts_ref_acc / nts_ref_norm_degree
 *       the primitives of core/ntsBaseOp.hpp:82-126,194-197, exactly as the
 *       reference compiles them (AVX path included).
 *   nts_ref_fused_forward / nts_ref_fused_backward
 *       ForwardCPUfuseOp::forward/backward (core/ntsCPUFusedGraphOp.hpp:41-167)
 *       run end to end at 1 rank: the reference's emit -> record -> sparse_slot
 *       CSC accumulate, and the CSR pull -> emit -> nts_acc merge.
 *   nts_ref_src_scatter_{fwd,bwd} / nts_ref_dst_aggregate_{fwd,bwd}
 *       SingleCPUSrcScatterOp / SingleCPUDstAggregateOp
 *       (core/ntsSingleCPUGraphOp.hpp:94-204), the edge-decomposed (GAT-path)
 *       forms.
 *
 * This library is the "reference-executed" parity anchor:
 * only tests/, tests/golden/make_golden.py and bench.py's cpu_baseline may
 * load it.  It is built in the dev container (where /root/reference exists)
 * and travels to the GPU box as a prebuilt .so (gitignored, not
 * gpurun-ignored).
 */
#include <cstdint>
#include <cstring>
#include <vector>

#include "core/ntsCPUFusedGraphOp.hpp"   /* reference source, via -I */
/* ntsMiniBatchGraphOp.hpp includes "ntsSampler.hpp", which resolves to the
 * includer's directory (the REAL core/ntsSampler.hpp -> FullyRepGraph/MPI
 * machinery).  Neutralize its guard and supply the 1-rank stub surface
 * instead; the op's loop bodies still compile from the reference tree. */
#define NTSSAMPLER_HPP
#include "ntsSampler.hpp"                /* stub/ntsSampler.hpp via -Istub */
#include "core/ntsMiniBatchGraphOp.hpp"  /* reference source, via -I */
#include "core/ntsDistCPUGraphOp.hpp"    /* reference source, via -I */
#include "core/ntsSingleCPUGraphOp.hpp"  /* reference source, via -I */

using nts::op::ForwardCPUfuseOp;
using nts::op::MiniBatchFuseOp;
using nts::op::SingleCPUDstAggregateOp;
using nts::op::SingleCPUSrcScatterOp;

#define EXPORT extern "C" __attribute__((visibility("default")))

namespace {

/* Assemble the 1-rank stub world around caller-provided arrays. */
struct World {
  Graph<Empty> g;
  PartitionedGraph pg;
  CSC_segment_pinned chunk;
  StubGnnCtx ctx;
  StubRtmInfo rtm;
  StubNts nts;
  StubComm comm;
  VertexSubset active;
  VertexId offs[2];

  World(uint32_t v, uint64_t e, const uint32_t *col_off, const uint32_t *rows,
        const uint32_t *row_off, const uint32_t *cols, const uint32_t *outdeg,
        const uint32_t *indeg) {
    g.vertices = v;
    g.edges = e;
    offs[0] = 0;
    offs[1] = v;
    g.partition_offset = offs;
    g.local_partition_offset = offs;
    g.out_degree_for_backward = const_cast<uint32_t *>(outdeg);
    g.in_degree_for_backward = const_cast<uint32_t *>(indeg);
    ctx.p_v_s = 0;
    ctx.p_v_e = v;
    ctx.l_v_num = v;
    ctx.l_e_num = (VertexId)e;
    g.gnnctx = &ctx;
    rtm.lock_free = false;
    g.rtminfo = &rtm;
    g.Nts = &nts;
    comm.g = &g;
    g.NtsComm = &comm;
    chunk.column_offset = const_cast<uint32_t *>(col_off);
    chunk.row_indices = const_cast<uint32_t *>(rows);
    chunk.row_offset = const_cast<uint32_t *>(row_off);
    chunk.column_indices = const_cast<uint32_t *>(cols);
    pg.graph_ = &g;
    pg.graph_chunks.push_back(&chunk);
    active.start = 0;
    active.end = v;
  }
};

}  // namespace

/* ---- primitives (core/ntsBaseOp.hpp) ---- */

EXPORT void nts_ref_comp(float *output, const float *input, float weight,
                         int feat_size) {
  nts::op::nts_comp(output, const_cast<float *>(input), weight, feat_size);
}

EXPORT void nts_ref_acc(float *output, const float *input, int feat_size) {
  nts::op::nts_acc(output, const_cast<float *>(input), feat_size);
}

EXPORT float nts_ref_norm_degree(uint32_t src, uint32_t dst,
                                 const uint32_t *outdeg,
                                 const uint32_t *indeg, uint32_t v) {
  Graph<Empty> g;
  g.vertices = v;
  g.out_degree_for_backward = const_cast<uint32_t *>(outdeg);
  g.in_degree_for_backward = const_cast<uint32_t *>(indeg);
  return nts::op::nts_norm_degree(&g, src, dst);
}

/* ---- ForwardCPUfuseOp (core/ntsCPUFusedGraphOp.hpp:41-167) ---- */

EXPORT void nts_ref_fused_forward(uint32_t v, int64_t f,
                                  const uint32_t *col_off, const uint32_t *rows,
                                  const uint32_t *row_off, const uint32_t *cols,
                                  const uint32_t *outdeg, const uint32_t *indeg,
                                  const float *x, float *y) {
  World w(v, col_off[v], col_off, rows, row_off, cols, outdeg, indeg);
  ForwardCPUfuseOp op(&w.pg, &w.active);
  NtsVar xv(v, f, const_cast<float *>(x));
  NtsVar yv = op.forward(xv);
  memcpy(y, yv.data(), sizeof(float) * v * f);
}

EXPORT void nts_ref_fused_backward(uint32_t v, int64_t f,
                                   const uint32_t *col_off, const uint32_t *rows,
                                   const uint32_t *row_off, const uint32_t *cols,
                                   const uint32_t *outdeg, const uint32_t *indeg,
                                   const float *gy, float *gx) {
  World w(v, col_off[v], col_off, rows, row_off, cols, outdeg, indeg);
  ForwardCPUfuseOp op(&w.pg, &w.active);
  NtsVar gv(v, f, const_cast<float *>(gy));
  NtsVar gxv = op.backward(gv);
  memcpy(gx, gxv.data(), sizeof(float) * v * f);
}

/* ---- SingleCPUSrcScatterOp (core/ntsSingleCPUGraphOp.hpp:94-147) ---- */

EXPORT void nts_ref_src_scatter_fwd(uint32_t v, uint64_t e, int64_t f,
                                    const uint32_t *col_off,
                                    const uint32_t *rows, const float *x,
                                    float *msg) {
  World w(v, e, col_off, rows, nullptr, nullptr, nullptr, nullptr);
  SingleCPUSrcScatterOp op(&w.pg, &w.active);
  NtsVar xv(v, f, const_cast<float *>(x));
  NtsVar mv = op.forward(xv);
  memcpy(msg, mv.data(), sizeof(float) * e * f);
}

EXPORT void nts_ref_src_scatter_bwd(uint32_t v, uint64_t e, int64_t f,
                                    const uint32_t *col_off,
                                    const uint32_t *rows, const float *msg_grad,
                                    float *x_grad) {
  World w(v, e, col_off, rows, nullptr, nullptr, nullptr, nullptr);
  SingleCPUSrcScatterOp op(&w.pg, &w.active);
  NtsVar gv(e, f, const_cast<float *>(msg_grad));
  NtsVar xv = op.backward(gv);
  memcpy(x_grad, xv.data(), sizeof(float) * v * f);
}

/* ---- SingleCPUDstAggregateOp (core/ntsSingleCPUGraphOp.hpp:149-204) ---- */

EXPORT void nts_ref_dst_aggregate_fwd(uint32_t v, uint64_t e, int64_t f,
                                      const uint32_t *col_off,
                                      const uint32_t *rows, const float *msg,
                                      float *y) {
  World w(v, e, col_off, rows, nullptr, nullptr, nullptr, nullptr);
  SingleCPUDstAggregateOp op(&w.pg, &w.active);
  NtsVar mv(e, f, const_cast<float *>(msg));
  NtsVar yv = op.forward(mv);
  memcpy(y, yv.data(), sizeof(float) * v * f);
}

EXPORT void nts_ref_dst_aggregate_bwd(uint32_t v, uint64_t e, int64_t f,
                                      const uint32_t *col_off,
                                      const uint32_t *rows, const float *y_grad,
                                      float *msg_grad) {
  World w(v, e, col_off, rows, nullptr, nullptr, nullptr, nullptr);
  SingleCPUDstAggregateOp op(&w.pg, &w.active);
  NtsVar gv(v, f, const_cast<float *>(y_grad));
  NtsVar mv = op.backward(gv);
  memcpy(msg_grad, mv.data(), sizeof(float) * e * f);
}

/* ---- MiniBatchFuseOp (core/ntsMiniBatchGraphOp.hpp:61-131) ----
 * Sampled-subgraph aggregation: local CSC over n_dst sampled destinations,
 * r_i entries are LOCAL source slots, dst_ids/src_ids map locals to global
 * vertex ids (for the norm-degree weights). */
static void minibatch_world(Graph<Empty> &g, StubGnnCtx &ctx, StubNts &nts,
                            const uint32_t *outdeg, const uint32_t *indeg,
                            uint32_t v) {
  g.vertices = v;
  g.out_degree_for_backward = const_cast<uint32_t *>(outdeg);
  g.in_degree_for_backward = const_cast<uint32_t *>(indeg);
  ctx.p_v_s = 0;
  g.gnnctx = &ctx;
  g.Nts = &nts;
}

EXPORT void nts_ref_minibatch_forward(uint32_t v, uint32_t n_dst,
                                      uint32_t n_src, int64_t f,
                                      const uint32_t *col_off,
                                      const uint32_t *r_i_local,
                                      const uint32_t *dst_ids,
                                      const uint32_t *src_ids,
                                      const uint32_t *outdeg,
                                      const uint32_t *indeg, const float *x,
                                      float *y) {
  Graph<Empty> g;
  StubGnnCtx ctx;
  StubNts nts;
  minibatch_world(g, ctx, nts, outdeg, indeg, v);
  sampCSC sg;
  sg.v_dst.assign(dst_ids, dst_ids + n_dst);
  sg.v_src.assign(src_ids, src_ids + n_src);
  sg.c_o.assign(col_off, col_off + n_dst + 1);
  sg.r_indices.assign(r_i_local, r_i_local + col_off[n_dst]);
  SampledSubgraph ssg;
  ssg.sampled_sgs.push_back(&sg);
  MiniBatchFuseOp op(&ssg, &g, /*layer=*/0);
  NtsVar xv(n_src, f, const_cast<float *>(x));
  NtsVar yv = op.forward(xv);
  memcpy(y, yv.data(), sizeof(float) * n_dst * f);
}

EXPORT void nts_ref_minibatch_backward(uint32_t v, uint32_t n_dst,
                                       uint32_t n_src, int64_t f,
                                       const uint32_t *col_off,
                                       const uint32_t *r_i_local,
                                       const uint32_t *dst_ids,
                                       const uint32_t *src_ids,
                                       const uint32_t *outdeg,
                                       const uint32_t *indeg, const float *gy,
                                       float *gx) {
  Graph<Empty> g;
  StubGnnCtx ctx;
  StubNts nts;
  minibatch_world(g, ctx, nts, outdeg, indeg, v);
  sampCSC sg;
  sg.v_dst.assign(dst_ids, dst_ids + n_dst);
  sg.v_src.assign(src_ids, src_ids + n_src);
  sg.c_o.assign(col_off, col_off + n_dst + 1);
  sg.r_indices.assign(r_i_local, r_i_local + col_off[n_dst]);
  SampledSubgraph ssg;
  ssg.sampled_sgs.push_back(&sg);
  MiniBatchFuseOp op(&ssg, &g, 0);
  NtsVar gv(n_dst, f, const_cast<float *>(gy));
  NtsVar gxv = op.backward(gv);
  memcpy(gx, gxv.data(), sizeof(float) * n_src * f);
}

/* ---- dist-GAT mirror ops (core/ntsDistCPUGraphOp.hpp) ----
 * The whole-dst-range CSC + compressed MirrorIndex machinery that
 * neutronstarlite_amd/dist_gat.py restates: DistGetDepNbrOp's
 * master->mirror gather / mirror->master grad return (:34-126) and the
 * MirrorIndex-indirected per-edge scatter/aggregate (:127-310), run at
 * 1 rank.  MirrorIndex is built HERE exactly as generateMirrorIndex does
 * (PartitionedGraph.hpp:295-305) from the chunk's row_indices. */
namespace {
struct DistWorld {
  Graph<Empty> g;
  PartitionedGraph pg;
  StubGnnCtx ctx;
  StubRtmInfo rtm;
  StubNts nts;
  StubComm comm;
  VertexSubset active;
  VertexId offs[2];
  std::vector<VertexId> mirror_index;
  CSC_segment_pinned chunk; /* carries the src-active filter the real
                               dep-neighbor driver applies */

  DistWorld(uint32_t v, uint32_t e, const uint32_t *col_off,
            const uint32_t *rows) {
    g.vertices = v;
    g.edges = e;
    offs[0] = 0;
    offs[1] = v;
    g.partition_offset = offs;
    g.local_partition_offset = offs;
    ctx.p_v_s = 0;
    ctx.p_v_e = v;
    ctx.l_v_num = v;
    ctx.l_e_num = e;
    g.gnnctx = &ctx;
    rtm.lock_free = false;
    g.rtminfo = &rtm;
    g.Nts = &nts;
    comm.g = &g;
    g.NtsComm = &comm;
    pg.graph_ = &g;
    pg.owned_vertices = v;
    pg.owned_edges = e;
    pg.column_offset = const_cast<uint32_t *>(col_off);
    pg.row_indices = const_cast<uint32_t *>(rows);
    /* generateMirrorIndex (PartitionedGraph.hpp:295-305), verbatim
     * semantics: prefix sum over "appears as src of an owned edge" */
    mirror_index.assign(v + 1, 0);
    for (VertexId i = 0; i < e; i++) mirror_index[rows[i] + 1] = 1;
    for (VertexId i = 0; i < v; i++) mirror_index[i + 1] += mirror_index[i];
    pg.owned_mirrors = mirror_index[v];
    pg.MirrorIndex = mirror_index.data();
    chunk.mirror_prefix = mirror_index.data();
    pg.graph_chunks.push_back(&chunk);
    active.start = 0;
    active.end = v;
  }
};
}  // namespace

EXPORT uint32_t nts_ref_dist_mirror_index(uint32_t v, uint32_t e,
                                          const uint32_t *col_off,
                                          const uint32_t *rows,
                                          uint32_t *out_index) {
  DistWorld w(v, e, col_off, rows);
  memcpy(out_index, w.pg.MirrorIndex, sizeof(uint32_t) * (v + 1));
  return w.pg.owned_mirrors;
}

EXPORT void nts_ref_dist_get_dep_nbr_fwd(uint32_t v, uint32_t e, int64_t f,
                                         const uint32_t *col_off,
                                         const uint32_t *rows, const float *x,
                                         float *mirror_out) {
  DistWorld w(v, e, col_off, rows);
  nts::op::DistGetDepNbrOp op(&w.pg, &w.active);
  NtsVar xv(v, f, const_cast<float *>(x));
  NtsVar mv = op.forward(xv);
  memcpy(mirror_out, mv.data(), sizeof(float) * w.pg.owned_mirrors * f);
}

EXPORT void nts_ref_dist_get_dep_nbr_bwd(uint32_t v, uint32_t e, int64_t f,
                                         const uint32_t *col_off,
                                         const uint32_t *rows,
                                         const float *mirror_grad,
                                         float *x_grad) {
  DistWorld w(v, e, col_off, rows);
  nts::op::DistGetDepNbrOp op(&w.pg, &w.active);
  /* harness guard: the op's backward reads row MirrorIndex[src] for EVERY
   * src, and vertices past the last mirror index one row past
   * owned_mirrors (a reference quirk — those vertices' grads come from an
   * out-of-bounds read there).  One zero guard row keeps the harness
   * well-defined: mirror-less vertices receive exactly 0, which is also
   * our implementation's semantics. */
  NtsVar gv(w.pg.owned_mirrors + 1, f);
  memcpy(gv.data(), mirror_grad,
         sizeof(float) * w.pg.owned_mirrors * f);
  NtsVar xv = op.backward(gv);
  memcpy(x_grad, xv.data(), sizeof(float) * v * f);
}

EXPORT void nts_ref_dist_scatter_src_fwd(uint32_t v, uint32_t e, int64_t f,
                                         const uint32_t *col_off,
                                         const uint32_t *rows,
                                         const float *mirror, float *msg) {
  DistWorld w(v, e, col_off, rows);
  nts::op::DistScatterSrc op(&w.pg, &w.active);
  NtsVar mv(w.pg.owned_mirrors, f, const_cast<float *>(mirror));
  NtsVar out = op.forward(mv);
  memcpy(msg, out.data(), sizeof(float) * e * f);
}

EXPORT void nts_ref_dist_scatter_src_bwd(uint32_t v, uint32_t e, int64_t f,
                                         const uint32_t *col_off,
                                         const uint32_t *rows,
                                         const float *msg_grad,
                                         float *mirror_grad) {
  DistWorld w(v, e, col_off, rows);
  nts::op::DistScatterSrc op(&w.pg, &w.active);
  NtsVar gv(e, f, const_cast<float *>(msg_grad));
  NtsVar out = op.backward(gv);
  memcpy(mirror_grad, out.data(), sizeof(float) * w.pg.owned_mirrors * f);
}

EXPORT void nts_ref_dist_aggregate_dst_fwd(uint32_t v, uint32_t e, int64_t f,
                                           const uint32_t *col_off,
                                           const uint32_t *rows,
                                           const float *msg, float *y) {
  DistWorld w(v, e, col_off, rows);
  nts::op::DistAggregateDst op(&w.pg, &w.active);
  NtsVar mv(e, f, const_cast<float *>(msg));
  NtsVar out = op.forward(mv);
  memcpy(y, out.data(), sizeof(float) * v * f);
}

EXPORT void nts_ref_dist_aggregate_dst_bwd(uint32_t v, uint32_t e, int64_t f,
                                           const uint32_t *col_off,
                                           const uint32_t *rows,
                                           const float *y_grad,
                                           float *msg_grad) {
  DistWorld w(v, e, col_off, rows);
  nts::op::DistAggregateDst op(&w.pg, &w.active);
  NtsVar gv(v, f, const_cast<float *>(y_grad));
  NtsVar out = op.backward(gv);
  memcpy(msg_grad, out.data(), sizeof(float) * e * f);
}

EXPORT int nts_ref_ok(void) { return 1; }
