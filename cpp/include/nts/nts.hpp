/* nts.hpp — C++ host layer: the reference's operator surface over libtorch,
 * computing through the C-ABI HIP shim (include/nts_hip.h).
 *
 * This keeps the calling convention of /root/reference's operator boundary
 *  so model code written against it — e.g. the per-layer loop
 * of toolkits/GCN.hpp:217-235 — compiles against this header:
 *   - NtsVar = torch::Tensor (core/NtsScheduler.hpp:52)
 *   - nts::op::ntsGraphOp: ctor (PartitionedGraph*, VertexSubset*),
 *     NtsVar forward(NtsVar&), NtsVar backward(NtsVar&)
 *     (core/ntsBaseOp.hpp:24-48)
 *   - class names ForwardSingleGPUfuseOp / ForwardGPUfuseOp
 *     (core/ntsSingleGPUFusedGraphOp.hpp:48-71, ntsDistGPUFusedGraphOp.hpp:48-91)
 *   - NtsContext::runGraphOp<T> / appendNNOp / self_backward tape
 *     (core/ntsContext.hpp:108-359)
 * The implementations are new (MI355X-first; no host bounce, no MPI): only
 * the surface matches.  Error culture: assert/abort, no exceptions
 * (reference convention).
 */
#pragma once

#include <torch/torch.h>

#include <cassert>
#include <cstdint>
#include <functional>
#include <vector>

#include "nts_comm.h"
#include "nts_hip.h"

namespace nts {

using VertexId = uint32_t;   /* dep/gemini/type.hpp:28 */
using ValueType = float;     /* dep/gemini/type.hpp:30 */
using NtsVar = torch::Tensor;

/* Active-vertex subset: carried for signature compatibility (full-batch
 * configs run with all vertices active). */
struct VertexSubset {
  VertexId start = 0, end = 0;
};

/* One per-source-partition graph chunk resident in HBM: forward CSC +
 * backward CSR + norm-degree weights (CSC_segment_pinned surface,
 * core/GraphSegment.h:52-139; device upload ≙ CopyGraphToDevice,
 * core/GraphSegment.cpp:178-220). */
struct CSC_segment_pinned {
  VertexId src_range[2] = {0, 0};
  VertexId dst_range[2] = {0, 0};
  VertexId edge_size = 0;
  torch::Tensor column_offset;   // u32 as int32, device, [dst_n+1] local
  torch::Tensor row_indices;     // global src ids, [E]
  torch::Tensor edge_weight_forward;
  torch::Tensor row_offset;      // [src_n+1] local
  torch::Tensor column_indices;  // global dst ids, [E]
  torch::Tensor edge_weight_backward;

  VertexId dst_n() const { return dst_range[1] - dst_range[0]; }
  VertexId src_n() const { return src_range[1] - src_range[0]; }

  /* Build from host CSC+CSR arrays (u32 / f32), upload to `device`. */
  static CSC_segment_pinned from_host(
      VertexId src_s, VertexId src_e, VertexId dst_s, VertexId dst_e,
      const uint32_t *col_off, const uint32_t *rows, const float *wf,
      const uint32_t *row_off, const uint32_t *cols, const float *wb,
      VertexId edges, torch::Device device) {
    CSC_segment_pinned c;
    c.src_range[0] = src_s; c.src_range[1] = src_e;
    c.dst_range[0] = dst_s; c.dst_range[1] = dst_e;
    c.edge_size = edges;
    auto u32 = torch::TensorOptions().dtype(torch::kInt32);
    auto f32 = torch::TensorOptions().dtype(torch::kFloat32);
    auto up = [&](const void *p, int64_t n, torch::TensorOptions o) {
      return torch::from_blob(const_cast<void *>(p), {n}, o).to(device);
    };
    c.column_offset = up(col_off, (int64_t)(dst_e - dst_s) + 1, u32);
    c.row_indices = up(rows, edges, u32);
    c.edge_weight_forward = up(wf, edges, f32);
    c.row_offset = up(row_off, (int64_t)(src_e - src_s) + 1, u32);
    c.column_indices = up(cols, edges, f32.dtype(torch::kInt32));
    c.edge_weight_backward = up(wb, edges, f32);
    return c;
  }
};

/* Partitioned-graph view: per-source-partition chunks of THIS rank
 * (PartitionedGraph surface, core/PartitionedGraph.hpp). */
struct PartitionedGraph {
  std::vector<CSC_segment_pinned *> graph_chunks;
  int partition_id = 0;
  std::vector<VertexId> partition_offset;  // [P+1]
  nts_stream *stream = nullptr;            // compute stream (C-ABI)
  nts_stream *comm_stream = nullptr;       // ring-exchange stream (overlap)
  nts_comm *comm = nullptr;                // RCCL ring (nullptr => P==1 only)
  torch::Tensor mirror_index;              // per-graph mirror compression
                                           // (generateMirrorIndex surface,
                                           // PartitionedGraph.hpp:295-305)

  /* Wrap the HIP null stream (= torch's default stream): libtorch tensor
   * fills/copies and our kernels must share one stream order. */
  PartitionedGraph() {
    stream = nts_stream_wrap(nullptr);
    /* the side stream needs a device; CPU-only host-logic tests
     * (cpp/host_unit_check.cpp) construct this object without one */
    if (nts_device_count() > 0)
      comm_stream = nts_stream_create();  /* non-blocking side stream */
  }
  ~PartitionedGraph() {
    if (stream) nts_stream_destroy(stream);
    if (comm_stream) nts_stream_destroy(comm_stream);
  }
  VertexId owned_vertices() const {
    return partition_offset[partition_id + 1] - partition_offset[partition_id];
  }
  int partitions() const { return (int)partition_offset.size() - 1; }
  VertexId part_n(int k) const {
    return partition_offset[k + 1] - partition_offset[k];
  }

  /* Single-partition mirror index = identity over the whole graph (the
   * multi-partition compressed index is built at partition time).  Stored
   * per graph object — NOT function-static — so several graphs/devices can
   * coexist in one process. */
  torch::Tensor &mirror_index_for(torch::Device dev) {
    const int64_t v = partition_offset.back();
    if (!mirror_index.defined() || mirror_index.size(0) != v ||
        mirror_index.device() != dev) {
      mirror_index = torch::arange(
          v, torch::TensorOptions().dtype(torch::kInt32).device(dev));
    }
    return mirror_index;
  }
};

namespace op {

/* Abstract graph-op (core/ntsBaseOp.hpp:24-48 surface). */
class ntsGraphOp {
 public:
  PartitionedGraph *partitioned_graph_ = nullptr;
  VertexSubset *active_ = nullptr;
  ntsGraphOp() {}
  ntsGraphOp(PartitionedGraph *pg, VertexSubset *active)
      : partitioned_graph_(pg), active_(active) {}
  virtual ~ntsGraphOp() {}
  virtual NtsVar forward(NtsVar &f_input) = 0;
  virtual NtsVar backward(NtsVar &output_grad) = 0;
};

namespace detail {

inline void csc_forward(nts_stream *s, const CSC_segment_pinned &c,
                        const NtsVar &x, NtsVar &y) {
  assert(x.is_cuda() && x.is_contiguous() &&
         x.size(0) == (int64_t)c.src_n());
  nts_gather_by_dst_from_src(
      s, x.data_ptr<float>(), y.data_ptr<float>(),
      c.edge_weight_forward.data_ptr<float>(),
      (const uint32_t *)c.row_indices.data_ptr<int32_t>(),
      (const uint32_t *)c.column_offset.data_ptr<int32_t>(),
      c.src_range[0], c.src_range[1], c.dst_range[0], c.dst_range[1],
      c.edge_size, c.dst_n(), (uint32_t)x.size(1), 1);
}

inline void csr_backward(nts_stream *s, const CSC_segment_pinned &c,
                         const NtsVar &g, NtsVar &out) {
  assert(g.is_cuda() && g.is_contiguous() &&
         g.size(0) == (int64_t)c.dst_n());
  nts_gather_by_src_from_dst(
      s, g.data_ptr<float>(), out.data_ptr<float>(),
      c.edge_weight_backward.data_ptr<float>(),
      (const uint32_t *)c.row_offset.data_ptr<int32_t>(),
      (const uint32_t *)c.column_indices.data_ptr<int32_t>(),
      c.src_range[0], c.src_range[1], c.dst_range[0], c.dst_range[1],
      c.edge_size, c.src_n(), (uint32_t)g.size(1), 1);
}

}  // namespace detail

/* Single-GPU fused aggregation op (ForwardSingleGPUfuseOp surface,
 * core/ntsSingleGPUFusedGraphOp.hpp:48-71): the whole graph is one chunk. */
class ForwardSingleGPUfuseOp : public ntsGraphOp {
 public:
  ForwardSingleGPUfuseOp(PartitionedGraph *pg, VertexSubset *active)
      : ntsGraphOp(pg, active) {}
  NtsVar forward(NtsVar &f_input) override {
    auto &c = *partitioned_graph_->graph_chunks[0];
    NtsVar y = torch::zeros({(int64_t)c.dst_n(), f_input.size(1)},
                            f_input.options());
    detail::csc_forward(partitioned_graph_->stream, c, f_input, y);
    return y;
  }
  NtsVar backward(NtsVar &output_grad) override {
    auto &c = *partitioned_graph_->graph_chunks[0];
    NtsVar gx = torch::zeros({(int64_t)c.src_n(), output_grad.size(1)},
                             output_grad.options());
    detail::csr_backward(partitioned_graph_->stream, c, output_grad, gx);
    return gx;
  }
};

/* Distributed fused aggregation op (ForwardGPUfuseOp surface,
 * core/ntsDistGPUFusedGraphOp.hpp:48-91) — THE named hot-path operator, so
 * toolkits/GCN.hpp:227 compiles unchanged against this layer.
 *
 * forward  ≙ Graph::sync_compute_decoupled (core/graph.hpp:3640-3719):
 *   the P-step master→mirror ring.  Each rank owns the contiguous vertex
 *   range partition_offset[r]..[r+1]; step s sends the rank's OWNED dense
 *   feature block to rank (r−s)%P and receives rank (r+s)%P's block, then
 *   aggregates that rank's chunk into the owned output.  Unlike the
 *   reference there is no f_input.cpu() bounce (:58), no [vid|f×f32]
 *   record packing (comm/network.cpp:476-495), no pinned-host spin queues:
 *   dense fp32 blocks move GPU→GPU by grouped ncclSend/ncclRecv over xGMI,
 *   stream-ordered with the aggregation kernels.
 * backward ≙ Graph::compute_sync_decoupled (core/graph.hpp:3456-3622):
 *   per remote partition k, the local chunk's CSR produces the partial
 *   gradient block for k's masters; ring-send it to its owner, who
 *   dense-adds (the reference's aggregate_data_buffer_debug merge,
 *   cuda/ntsCUDATransferKernel.cuh:49-68, becomes a dense +=).
 *
 * P==1 degenerates to the local chunk only (no comm calls), matching the
 * reference's single-rank pass-through (network.cpp:461-463).  The Python
 * driver (neutronstarlite_amd/ring.py) implements the same protocol over
 * torch.distributed with the overlap/pipelining variants; this C++ op is
 * the link-contract implementation over the nts_comm C-ABI. */
class ForwardGPUfuseOp : public ntsGraphOp {
 public:
  std::vector<CSC_segment_pinned *> subgraphs;
  ForwardGPUfuseOp(PartitionedGraph *pg, VertexSubset *active)
      : ntsGraphOp(pg, active) {
    subgraphs = pg->graph_chunks;
  }
  NtsVar forward(NtsVar &f_input) override {
    auto *pg = partitioned_graph_;
    const int P = pg->partitions(), r = pg->partition_id;
    const int64_t f = f_input.size(1);
    assert(f_input.size(0) == (int64_t)pg->owned_vertices());
    assert(P == 1 || pg->comm);
    NtsVar x = f_input.contiguous();
    NtsVar y = torch::zeros({(int64_t)pg->owned_vertices(), f}, x.options());
    /* overlapped ring (PROC_OVERLAP default, graph.hpp:3490-3535): the
     * exchange for step s+1 runs on comm_stream while step s's block
     * aggregates on the compute stream.  Buffer safety across streams
     * relies on the wait points below: a tensor freed on the compute
     * stream is only REUSED by compute-stream work, which the waits order
     * after the comm stream finished touching it. */
    auto post = [&](int step, NtsVar &recv) {
      const int to = (r - step + P) % P, frm = (r + step) % P;
      recv = torch::empty({(int64_t)pg->part_n(frm), f}, x.options());
      nts_comm_group_begin();
      nts_comm_send_f32(pg->comm, pg->comm_stream, x.data_ptr<float>(),
                        x.numel(), to);
      nts_comm_recv_f32(pg->comm, pg->comm_stream, recv.data_ptr<float>(),
                        recv.numel(), frm);
      nts_comm_group_end();
    };
    NtsVar bufs[2];
    if (P > 1) {
      /* x is produced on the compute stream; one dependency covers every
       * later send of the same buffer */
      nts_stream_wait_stream(pg->comm_stream, pg->stream);
      post(1, bufs[1 % 2]);
    }
    detail::csc_forward(pg->stream, *subgraphs[r], x, y);
    for (int step = 1; step < P; step++) {
      /* comm tail == exchange `step` here (step+1 not yet posted), so
       * this wait releases exactly when our block has arrived */
      nts_stream_wait_stream(pg->stream, pg->comm_stream);
      if (step + 1 < P) post(step + 1, bufs[(step + 1) % 2]);
      detail::csc_forward(pg->stream, *subgraphs[(r + step) % P],
                          bufs[step % 2], y);
    }
    return y;
  }
  NtsVar backward(NtsVar &output_grad) override {
    auto *pg = partitioned_graph_;
    const int P = pg->partitions(), r = pg->partition_id;
    const int64_t f = output_grad.size(1);
    assert(output_grad.size(0) == (int64_t)pg->owned_vertices());
    assert(P == 1 || pg->comm);
    NtsVar g = output_grad.contiguous();
    NtsVar gx = torch::zeros({(int64_t)pg->owned_vertices(), f}, g.options());
    detail::csr_backward(pg->stream, *subgraphs[r], g, gx);
    /* pipelined: step s+1's partial computes while step s's exchange is
     * in flight (compute_sync_decoupled semantics, graph.hpp:3456-3622).
     * Per iteration: compute partial_s -> merge recv_{s-1} (the comm tail
     * is exchange s-1 at that point, so the wait releases exactly on its
     * arrival) -> post exchange_s. */
    NtsVar recvs[2], partials[2];
    for (int step = 1; step < P; step++) {
      const int k = (r + step) % P;            /* owner we feed */
      const int peer_src = (r - step + P) % P; /* partial arriving for us */
      NtsVar &partial = partials[step % 2];
      partial = torch::zeros({(int64_t)pg->part_n(k), f}, g.options());
      detail::csr_backward(pg->stream, *subgraphs[k], g, partial);
      if (step >= 2) {
        nts_stream_wait_stream(pg->stream, pg->comm_stream);
        gx += recvs[(step - 1) % 2];
      }
      nts_stream_wait_stream(pg->comm_stream, pg->stream); /* partial done */
      NtsVar &recv = recvs[step % 2];
      recv = torch::empty_like(gx);
      nts_comm_group_begin();
      nts_comm_send_f32(pg->comm, pg->comm_stream, partial.data_ptr<float>(),
                        partial.numel(), k);
      nts_comm_recv_f32(pg->comm, pg->comm_stream, recv.data_ptr<float>(),
                        recv.numel(), peer_src);
      nts_comm_group_end();
    }
    if (P > 1) {
      nts_stream_wait_stream(pg->stream, pg->comm_stream);
      gx += recvs[(P - 1) % 2];
    }
    return gx;
  }
};

/* ---- decomposed edge-valued ops (GAT path, config #5) ----
 * The class names and signatures of core/ntsDistGPUGraphOp.hpp:48-361, so
 * GAT_GPU_DIST.hpp's 5-op chain (:191-215) compiles unchanged.  This host
 * layer implements the single-partition case (every vertex its own master;
 * mirror_index = identity — PartitionedGraph::generateMirrorIndex,
 * PartitionedGraph.hpp:295-305); the multi-partition mirror exchange lives
 * in the RCCL ring driver. */

namespace detail {
inline torch::Tensor &identity_mirror_index(PartitionedGraph *pg,
                                            torch::Device dev) {
  return pg->mirror_index_for(dev);  /* per-graph state, not function-static */
}
}  // namespace detail

/* gather dependent-neighbor features into the mirror matrix: with one
 * partition this is the identity over the input (DistGPUGetDepNbrOp,
 * ntsDistGPUGraphOp.hpp:48-143 — comm path degenerates). */
class DistGPUGetDepNbrOp : public ntsGraphOp {
 public:
  DistGPUGetDepNbrOp(PartitionedGraph *pg, VertexSubset *active)
      : ntsGraphOp(pg, active) {}
  NtsVar forward(NtsVar &f_input) override { return f_input.clone(); }
  NtsVar backward(NtsVar &output_grad) override { return output_grad.clone(); }
};

/* per-edge materialize of source-mirror rows (DistGPUScatterSrc,
 * ntsDistGPUGraphOp.hpp:145-196). */
class DistGPUScatterSrc : public ntsGraphOp {
 public:
  DistGPUScatterSrc(PartitionedGraph *pg, VertexSubset *active)
      : ntsGraphOp(pg, active) {}
  NtsVar forward(NtsVar &mirror) override {
    auto &c = *partitioned_graph_->graph_chunks[0];
    auto &mi = detail::identity_mirror_index(partitioned_graph_,
                                             mirror.device());
    NtsVar msg = torch::zeros({(int64_t)c.edge_size, mirror.size(1)},
                              mirror.options());
    nts_scatter_src_mirror_to_msg(
        partitioned_graph_->stream, msg.data_ptr<float>(),
        mirror.contiguous().data_ptr<float>(),
        (const uint32_t *)c.row_indices.data_ptr<int32_t>(),
        (const uint32_t *)c.column_offset.data_ptr<int32_t>(),
        (const uint32_t *)mi.data_ptr<int32_t>(), c.dst_n(),
        (uint32_t)mirror.size(1));
    return msg;
  }
  NtsVar backward(NtsVar &msg_grad) override {
    auto &c = *partitioned_graph_->graph_chunks[0];
    auto &mi = detail::identity_mirror_index(partitioned_graph_,
                                             msg_grad.device());
    NtsVar g = torch::zeros({(int64_t)c.src_n(), msg_grad.size(1)},
                            msg_grad.options());
    nts_gather_msg_to_src_mirror(
        partitioned_graph_->stream, g.data_ptr<float>(),
        msg_grad.contiguous().data_ptr<float>(),
        (const uint32_t *)c.row_indices.data_ptr<int32_t>(),
        (const uint32_t *)c.column_offset.data_ptr<int32_t>(),
        (const uint32_t *)mi.data_ptr<int32_t>(), c.dst_n(),
        (uint32_t)msg_grad.size(1));
    return g;
  }
};

/* per-edge materialize of destination rows (DistGPUScatterDst,
 * ntsDistGPUGraphOp.hpp:198-248). */
class DistGPUScatterDst : public ntsGraphOp {
 public:
  DistGPUScatterDst(PartitionedGraph *pg, VertexSubset *active)
      : ntsGraphOp(pg, active) {}
  NtsVar forward(NtsVar &dst_feat) override {
    auto &c = *partitioned_graph_->graph_chunks[0];
    NtsVar msg = torch::zeros({(int64_t)c.edge_size, dst_feat.size(1)},
                              dst_feat.options());
    nts_scatter_dst_to_msg(
        partitioned_graph_->stream, msg.data_ptr<float>(),
        dst_feat.contiguous().data_ptr<float>(),
        (const uint32_t *)c.row_indices.data_ptr<int32_t>(),
        (const uint32_t *)c.column_offset.data_ptr<int32_t>(), c.dst_n(),
        (uint32_t)dst_feat.size(1));
    return msg;
  }
  NtsVar backward(NtsVar &msg_grad) override {
    auto &c = *partitioned_graph_->graph_chunks[0];
    NtsVar g = torch::zeros({(int64_t)c.dst_n(), msg_grad.size(1)},
                            msg_grad.options());
    nts_gather_msg_to_dst(
        partitioned_graph_->stream, g.data_ptr<float>(),
        msg_grad.contiguous().data_ptr<float>(),
        (const uint32_t *)c.row_indices.data_ptr<int32_t>(),
        (const uint32_t *)c.column_offset.data_ptr<int32_t>(), c.dst_n(),
        (uint32_t)msg_grad.size(1));
    return g;
  }
};

/* edge->destination reduce (DistGPUAggregateDst,
 * ntsDistGPUGraphOp.hpp:250-300). */
class DistGPUAggregateDst : public ntsGraphOp {
 public:
  DistGPUAggregateDst(PartitionedGraph *pg, VertexSubset *active)
      : ntsGraphOp(pg, active) {}
  NtsVar forward(NtsVar &msg) override {
    auto &c = *partitioned_graph_->graph_chunks[0];
    NtsVar y = torch::zeros({(int64_t)c.dst_n(), msg.size(1)},
                            msg.options());
    nts_gather_msg_to_dst(
        partitioned_graph_->stream, y.data_ptr<float>(),
        msg.contiguous().data_ptr<float>(),
        (const uint32_t *)c.row_indices.data_ptr<int32_t>(),
        (const uint32_t *)c.column_offset.data_ptr<int32_t>(), c.dst_n(),
        (uint32_t)msg.size(1));
    return y;
  }
  NtsVar backward(NtsVar &y_grad) override {
    auto &c = *partitioned_graph_->graph_chunks[0];
    NtsVar mg = torch::zeros({(int64_t)c.edge_size, y_grad.size(1)},
                             y_grad.options());
    nts_scatter_grad_back_to_message(
        partitioned_graph_->stream, y_grad.contiguous().data_ptr<float>(),
        mg.data_ptr<float>(),
        (const uint32_t *)c.row_indices.data_ptr<int32_t>(),
        (const uint32_t *)c.column_offset.data_ptr<int32_t>(), c.dst_n(),
        (uint32_t)y_grad.size(1));
    return mg;
  }
};

/* per-destination softmax over incident edge values (DistGPUEdgeSoftMax,
 * ntsDistGPUGraphOp.hpp:302-361; no max subtraction, cached for backward). */
class DistGPUEdgeSoftMax : public ntsGraphOp {
  NtsVar cached_;
 public:
  DistGPUEdgeSoftMax(PartitionedGraph *pg, VertexSubset *active)
      : ntsGraphOp(pg, active) {}
  NtsVar forward(NtsVar &msg) override {
    auto &c = *partitioned_graph_->graph_chunks[0];
    NtsVar out = torch::zeros_like(msg);
    cached_ = torch::zeros_like(msg);
    nts_edge_softmax_forward(
        partitioned_graph_->stream, out.data_ptr<float>(),
        msg.contiguous().data_ptr<float>(), cached_.data_ptr<float>(),
        (const uint32_t *)c.row_indices.data_ptr<int32_t>(),
        (const uint32_t *)c.column_offset.data_ptr<int32_t>(), c.dst_n(),
        (uint32_t)msg.size(1));
    return out;
  }
  NtsVar backward(NtsVar &out_grad) override {
    auto &c = *partitioned_graph_->graph_chunks[0];
    NtsVar g = torch::zeros_like(out_grad);
    nts_edge_softmax_backward(
        partitioned_graph_->stream, g.data_ptr<float>(),
        out_grad.contiguous().data_ptr<float>(),
        cached_.data_ptr<float>(),
        (const uint32_t *)c.row_indices.data_ptr<int32_t>(),
        (const uint32_t *)c.column_offset.data_ptr<int32_t>(), c.dst_n(),
        (uint32_t)out_grad.size(1));
    return g;
  }
};

}  // namespace op

/* Layer weights + hand-rolled Adam + gradient allreduce (Parameter
 * surface, core/NtsScheduler.hpp:639-791), so toolkits' Update() loops —
 * P[i]->all_reduce_to_gradient(...); P[i]->learnC2G_with_decay_Adam();
 * P[i]->next(); (toolkits/GCN.hpp:205-214) — compile unchanged.
 *
 * MI355X-native differences (surface identical, plumbing new): the
 * reference allreduces on HOST through Network_simple's MPI_Allreduce
 * (network.h:198-203) and keeps Adam state split CPU/GPU; here gradients
 * and Adam state stay in HBM and the allreduce is RCCL over xGMI
 * (nts_comm_allreduce_sum_f32) on the caller's stream — comm == nullptr
 * degenerates to single-rank (no-op allreduce), matching np=1. */
struct Parameter {
  NtsVar W;       /* weights (leaf, requires_grad) */
  NtsVar W_gradient, M_GPU, V_GPU, W_g;
  ValueType alpha = 0.01f, beta1 = 0.9f, beta2 = 0.999f, epsilon = 1e-9f;
  ValueType alpha_t, beta1_t, beta2_t;
  ValueType weight_decay = 0.f, decay_rate = 1.f;
  long decay_epoch = -1, curr_epoch = 0;
  long row, col;
  nts_comm *comm = nullptr;
  nts_stream *stream = nullptr;

  /* Xavier-uniform init, NtsScheduler.hpp:669-672 convention */
  Parameter(size_t w, size_t h, ValueType alpha_ = 0.01f,
            ValueType beta1_ = 0.9f, ValueType beta2_ = 0.999f,
            ValueType epsilon_ = 1e-9f, ValueType weight_decay_ = 0.f)
      : alpha(alpha_), beta1(beta1_), beta2(beta2_), epsilon(epsilon_),
        weight_decay(weight_decay_), row((long)w), col((long)h) {
    const ValueType scale = std::sqrt(6.0 / (w + h));
    W = ((2 * scale) * torch::rand({(long)w, (long)h}) - scale)
            .set_requires_grad(true);
    alpha_t = alpha; beta1_t = beta1; beta2_t = beta2;
  }

  void to(torch::Device dev) {
    W = W.detach().to(dev).set_requires_grad(true);
    Adam_to_GPU(dev);
  }
  void Adam_to_GPU(torch::Device dev) { /* NtsScheduler.hpp:754-758 */
    M_GPU = torch::zeros({row, col}, torch::device(dev));
    V_GPU = torch::zeros({row, col}, torch::device(dev));
  }

  /* rank-0 weight broadcast at init (NtsScheduler.hpp:716-718 ->
   * MPI_Bcast; here ncclBroadcast) */
  void init_parameter() {
    if (comm && W.is_cuda()) {
      NtsVar w = W.detach().contiguous();
      nts_comm_bcast_f32(comm, stream, w.data_ptr<float>(), w.numel(), 0);
    }
  }

  /* DDP gradient sum (NtsScheduler.hpp:719-722); accepts the grad on any
   * device — no .cpu() bounce is required (or useful) here */
  void all_reduce_to_gradient(NtsVar from) {
    W_gradient = from.contiguous();
    if (comm && W_gradient.is_cuda()) {
      nts_comm_allreduce_sum_f32(comm, stream, W_gradient.data_ptr<float>(),
                                 W_gradient.data_ptr<float>(),
                                 W_gradient.numel());
      nts_stream_sync(stream);
    }
  }

  void set_decay(ValueType decay_rate_, long decay_epoch_) {
    decay_rate = decay_rate_;
    decay_epoch = decay_epoch_;
  }

  /* epoch bookkeeping; keeps the reference's (idiosyncratic) running
   * bias-correction exactly (NtsScheduler.hpp:725-733) */
  void next() {
    if (decay_epoch != -1 && curr_epoch != 0 &&
        curr_epoch % decay_epoch == 0)
      alpha_t *= decay_rate;
    alpha = alpha_t * std::sqrt(1 - beta2) / (1 - beta1);
    beta1 *= beta1_t;
    beta2 *= beta2_t;
    curr_epoch++;
  }

  NtsVar forward(NtsVar x) { return x.mm(W); }

  /* Adam step with decoupled weight decay (NtsScheduler.hpp:759-767
   * semantics), entirely on device */
  void learnC2G_with_decay_Adam() {
    torch::NoGradGuard ng;
    NtsVar g = W_gradient.to(W.device()) + weight_decay * W.detach();
    M_GPU = beta1 * M_GPU + (1 - beta1) * g;
    V_GPU = beta2 * V_GPU + (1 - beta2) * g * g;
    W.set_data(W.detach() - alpha * M_GPU / (torch::sqrt(V_GPU) + epsilon));
    if (W.grad().defined()) W.mutable_grad().zero_();
  }
};

/* Tape-based context (NtsContext surface, core/ntsContext.hpp:108-359):
 * graph ops bypass libtorch autograd; NN segments use it.  self_backward
 * walks the tape mixing torch::autograd::grad with ntsGraphOp::backward. */
class NtsContext {
  enum Kind { GRAPHOP, NNOP };
  struct Entry {
    Kind kind;
    NtsVar input, output;
    op::ntsGraphOp *op = nullptr;
  };
  std::vector<Entry> tape_;

 public:
  ~NtsContext() { reset(); }

  template <typename GOPT>
  NtsVar runGraphOp(PartitionedGraph *pg, VertexSubset *active,
                    NtsVar &f_input) {
    auto *op = new GOPT(pg, active);
    NtsVar out = op->forward(f_input);
    tape_.push_back({GRAPHOP, f_input, out, op});
    return out;
  }

  /* vertexForward: an NN function applied under torch autograd
   * (ntsContext.hpp:198-226 semantics, consecutive NN ops chained). */
  NtsVar runVertexForward(std::function<NtsVar(NtsVar &)> f, NtsVar &input) {
    NtsVar in_leaf = input.detach().requires_grad_(true);
    NtsVar out = f(in_leaf);
    tape_.push_back({NNOP, in_leaf, out, nullptr});
    return out;
  }

  /* Two-input form used by toolkits/GCN.hpp:228-232: only nbr_input is
   * taped (the reference's appendNNOp takes nbr_input,
   * ntsContext.hpp:198-206); vtx_input contributes through torch autograd
   * reachability only. */
  NtsVar runVertexForward(std::function<NtsVar(NtsVar &, NtsVar &)> f,
                          NtsVar &nbr_input, NtsVar &vtx_input) {
    NtsVar in_leaf = nbr_input.detach().requires_grad_(true);
    NtsVar out = f(in_leaf, vtx_input);
    tape_.push_back({NNOP, in_leaf, out, nullptr});
    return out;
  }

  /* Walk the tape backward from `loss` (ntsContext.hpp:276-359). Returns the
   * gradient w.r.t. the first entry's input. */
  NtsVar self_backward(NtsVar &loss) {
    NtsVar grad;  // grad w.r.t. current entry's OUTPUT
    for (auto it = tape_.rbegin(); it != tape_.rend(); ++it) {
      if (it->kind == NNOP) {
        /* accumulate into every reachable leaf (layer weights) AND the
         * segment input, like torch::backward at ntsContext.hpp:283-301 */
        if (!grad.defined()) {
          loss.backward();
        } else {
          torch::autograd::backward({it->output}, {grad});
        }
        grad = it->input.grad();
      } else {
        assert(grad.defined());
        grad = grad.contiguous();
        grad = it->op->backward(grad);
      }
    }
    reset();
    return grad;
  }

  void reset() {
    for (auto &e : tape_)
      if (e.op) delete e.op;  /* context owns graph ops, ntsContext.hpp:266-275 */
    tape_.clear();
  }
  size_t tape_size() const { return tape_.size(); }
};

}  // namespace nts
