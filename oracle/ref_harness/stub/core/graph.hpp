/* TEST INFRASTRUCTURE ONLY — stub "core/graph.hpp" for oracle/_ref.
 *
 * This header shadows /root/reference/core/graph.hpp (5 531 lines of MPI +
 * libtorch + NUMA machinery that cannot compile in this image) so that the
 * reference's OWN hot-path arithmetic sources compile and execute unmodified
 * from where they lie under /root/reference:
 *
 *   - core/ntsBaseOp.hpp        (nts_comp AVX axpy :82-104, nts_acc :114-126,
 *                                nts_norm_degree :194-197, ntsGraphOp base)
 *   - core/ntsCPUFusedGraphOp.hpp (ForwardCPUfuseOp::forward/backward
 *                                loop bodies :41-167)
 *   - core/ntsSingleCPUGraphOp.hpp (SingleCPUSrcScatterOp :94-147,
 *                                SingleCPUDstAggregateOp :149-204)
 *   - dep/gemini/type.hpp, dep/gemini/atomic.hpp (included for real below)
 *
 * The stub supplies only the 1-rank plumbing those sources call into —
 * exactly the degenerate single-partition semantics the reference itself
 * has at mpiexec -np 1, where the ring is a local pass-through
 * (comm/network.cpp:461-463): emit_buffer appends [u32 vid | f x f32]
 * records (record stride sizeofM<ValueType>(f) = 4+4f, graph.hpp:2055-2057)
 * into one in-memory MessageBuffer that the sparse_slot phase then reads.
 *
 * Nothing from the reference is copied into this file; the reference's code
 * is COMPILED FROM ITS OWN TREE via -I/root/reference (see the Makefile).
 * The resulting libntsref.so is the "reference"-kind parity anchor
 * (SURVEY.md §8c): tests/golden fixtures are regenerated from it and the
 * hand-written oracle/oracle.c must match it bit-exactly.
 *
 * Only tests/, tests/golden/make_golden.py and bench.py's cpu_baseline may
 * load the resulting library; the product path never touches it.
 */
#ifndef NTS_REF_STUB_GRAPH_HPP
#define NTS_REF_STUB_GRAPH_HPP

#include <assert.h>
#include <math.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include <cmath>
#include <cstring>
#include <memory>
#include <vector>

/* Real reference headers (self-contained, no MPI/torch): */
#include "dep/gemini/type.hpp"    /* VertexId, ValueType, Empty, VertexIndex,
                                     BackVertexIndex, VertexAdjList */
#include "dep/gemini/atomic.hpp"  /* write_add / write_min / write_max (cas) */

#ifndef LOG_INFO
#define LOG_INFO(...)  do { fprintf(stderr, __VA_ARGS__); fprintf(stderr, "\n"); } while (0)
#endif

/* torch::DeviceType tokens as used by the compiled ops; no real torch. */
namespace torch {
enum DeviceType { CPU, CUDA };
}

/* Minimal dense fp32 tensor standing in for NtsVar = torch::Tensor
 * (core/NtsScheduler.hpp:52).  Only what the compiled ops use: size(d),
 * a contiguous buffer, zero-init (reference factories are torch::zeros,
 * NtsScheduler.hpp:377-440). */
struct NtsVar {
  std::shared_ptr<std::vector<ValueType>> buf;
  long n = 0, f = 0;
  NtsVar() {}
  NtsVar(long n_, long f_, ValueType *borrow = nullptr) : n(n_), f(f_) {
    buf = std::make_shared<std::vector<ValueType>>((size_t)n_ * f_, 0.0f);
    if (borrow) memcpy(buf->data(), borrow, sizeof(ValueType) * n_ * f_);
  }
  long size(int d) const { return d == 0 ? n : f; }
  ValueType *data() { return buf->data(); }

  /* Compile-only surface so the whole reference header parses.  These are
   * used only by ops we do NOT pin (SingleEdgeSoftMax computes through
   * torch's own softmax, core/ntsSingleCPUGraphOp.hpp:365 — torch
   * semantics, not the hot-path arithmetic); executing them here aborts. */
  NtsVar slice(int, long, long, long) const { abort(); }
  NtsVar softmax(int) const { abort(); }
  NtsVar t() const { abort(); }
  NtsVar mm(const NtsVar &) const { abort(); }
  NtsVar operator*(const NtsVar &) const { abort(); }
  NtsVar operator-(const NtsVar &) const { abort(); }
  NtsVar &operator[](long) { abort(); } /* get_label's per-row copy */
};

struct VertexSubset {
  VertexId start = 0, end = 0;
};

/* CSC_segment_pinned members the compiled ops touch
 * (core/GraphSegment.h:52-139 surface). */
struct CSC_segment_pinned {
  VertexId *column_offset = nullptr;  /* [dst_n+1] */
  VertexId *row_indices = nullptr;    /* [E] */
  VertexId *row_offset = nullptr;     /* [src_n+1] */
  VertexId *column_indices = nullptr; /* [E] */
  VertexId *forward_multisocket_message_index = nullptr;
  BackVertexIndex *backward_multisocket_message_index = nullptr;
  /* when set, src_get_active answers "is v a source of an owned edge"
   * from the MirrorIndex prefix sum — the filter the real dep-neighbor
   * driver applies before slotting a received record
   * (core/graph.hpp:2907-2911) */
  const VertexId *mirror_prefix = nullptr;
  bool src_get_active(VertexId v) {
    return mirror_prefix ? mirror_prefix[v + 1] > mirror_prefix[v] : true;
  }
  bool get_forward_active(VertexId) { return true; }
};

struct MessageBuffer {
  char *data = nullptr;
};

template <typename EdgeData> class Graph;

/* 1-rank communicator: emit_buffer appends one [vid | f x f32] record per
 * call in call order (non-lock-free path, comm/network.cpp:476-495 layout). */
struct StubComm {
  Graph<Empty> *g = nullptr;
  inline void emit_buffer(VertexId vid, ValueType *data, int f_size);
  inline void emit_buffer_lock_free(VertexId vid, ValueType *data,
                                    VertexId write_index, int f_size) {
    (void)write_index;
    emit_buffer(vid, data, f_size); /* never taken: rtminfo.lock_free=false */
  }
};

struct StubGnnCtx {
  VertexId p_v_s = 0, p_v_e = 0;
  VertexId l_v_num = 0, l_e_num = 0;
};
struct StubRtmInfo {
  bool lock_free = false;
};

/* Tensor factory stand-in (NtsScheduler surface; zero-filled like the
 * reference's torch::zeros factories). */
struct StubNts {
  NtsVar NewKeyTensor(NtsVar &mould, torch::DeviceType) {
    return NtsVar(mould.n, mould.f);
  }
  NtsVar NewLeafTensor(NtsVar &mould, torch::DeviceType) {
    return NtsVar(mould.n, mould.f);
  }
  NtsVar NewKeyTensor(std::vector<long> shape, torch::DeviceType) {
    return NtsVar(shape[0], shape[1]);
  }
  NtsVar NewLeafTensor(std::vector<long> shape, torch::DeviceType) {
    return NtsVar(shape[0], shape[1]);
  }
  NtsVar NewLeafKLongTensor(std::vector<long>) { abort(); } /* get_label
      only; never executed through this stub */
  ValueType *getWritableBuffer(NtsVar &v, torch::DeviceType) {
    return v.data();
  }
};

template <typename EdgeData> class Graph {
public:
  /* members the compiled ops read */
  VertexId vertices = 0;
  EdgeId edges = 0;
  int partition_id = 0;
  int threads = 1;
  VertexId *partition_offset = nullptr;       /* [2] = {0, V} */
  VertexId *local_partition_offset = nullptr; /* [2] = {0, V} */
  VertexId *out_degree_for_backward = nullptr;
  VertexId *in_degree_for_backward = nullptr;
  StubGnnCtx *gnnctx = nullptr;
  StubRtmInfo *rtminfo = nullptr;
  StubNts *Nts = nullptr;
  StubComm *NtsComm = nullptr;

  /* record stride, core/graph.hpp:2055-2057 */
  template <typename M> inline size_t sizeofM(int f_size) {
    return sizeof(VertexId) + sizeof(M) * f_size;
  }

  /* ---- 1-rank emit/record state ---- */
  std::vector<char> rec_bytes;
  std::vector<VertexIndex> src_index; /* [V]: where each src's record landed */
  VertexId rec_count = 0;
  int rec_f = 0;

  void begin_records(int f_size) {
    rec_bytes.clear();
    rec_count = 0;
    rec_f = f_size;
    src_index.assign(vertices, VertexIndex{0, 0});
  }
  void append_record(VertexId vid, const ValueType *row, int f_size) {
    size_t stride = sizeofM<ValueType>(f_size);
    size_t at = rec_bytes.size();
    rec_bytes.resize(at + stride);
    memcpy(rec_bytes.data() + at, &vid, sizeof(VertexId));
    memcpy(rec_bytes.data() + at + sizeof(VertexId), row,
           sizeof(ValueType) * f_size);
    src_index[vid] = VertexIndex{0, rec_count++};
  }

  /* Forward driver (1-rank semantics of
   * process_edges_forward_decoupled_mutisockets, core/graph.hpp:2644):
   * signal every owned vertex (ascending), then slot every owned dst against
   * the single local chunk; the "received" buffer is the local send buffer
   * (the np=1 pass-through of comm/network.cpp:461-463). */
  template <typename R, typename M, typename SIG, typename SLOT>
  void process_edges_forward_decoupled_mutisockets(
      SIG sparse_signal, SLOT sparse_slot,
      std::vector<CSC_segment_pinned *> &subgraphs, int feature_size,
      VertexSubset *active) {
    (void)active;
    begin_records(feature_size);
    for (VertexId src = 0; src < vertices; src++)
      sparse_signal(src, /*current_send_partition=*/0);
    MessageBuffer mb;
    mb.data = rec_bytes.data();
    MessageBuffer *bufs[1] = {&mb};
    for (VertexId dst = 0; dst < vertices; dst++)
      sparse_slot(dst, subgraphs[0], bufs, src_index, /*recv_id=*/0);
  }

  /* Backward driver (1-rank semantics of
   * process_edges_backward_decoupled_multisockets, core/graph.hpp:3123):
   * per-vertex pull lambda on socket 0 / thread 0, then deliver each emitted
   * record to the msg lambda. */
  template <typename R, typename M, typename VTX, typename MSG>
  void process_edges_backward_decoupled_multisockets(VTX per_vertex,
                                                     MSG per_msg,
                                                     int feature_size,
                                                     VertexSubset *active) {
    (void)active;
    begin_records(feature_size);
    for (VertexId src = 0; src < vertices; src++)
      per_vertex(src, VertexAdjList<Empty>(), /*thread_id=*/0, /*recv_id=*/0,
                 /*socketId=*/0);
    size_t stride = sizeofM<ValueType>(feature_size);
    for (VertexId k = 0; k < rec_count; k++) {
      char *rec = rec_bytes.data() + (size_t)k * stride;
      VertexId vid;
      memcpy(&vid, rec, sizeof(VertexId));
      per_msg(vid, (ValueType *)(rec + sizeof(VertexId)));
    }
  }

  /* Dep-neighbor gather driver (get_from_dep_neighbor_mutisockets,
   * core/graph.hpp surface) at 1 rank: every master emits its row
   * (ascending), then the slot lambda consumes each record in arrival
   * (= emit) order — the ascending order is load-bearing for the
   * reference's MirrorIndex clobber-then-correct behavior on non-mirror
   * vertices. */
  template <typename R, typename M, typename SIG, typename SLOT>
  void get_from_dep_neighbor_mutisockets(SIG sparse_signal, SLOT sparse_slot,
                                         std::vector<CSC_segment_pinned *> &subgraphs,
                                         int feature_size,
                                         VertexSubset *active) {
    (void)active;
    begin_records(feature_size);
    for (VertexId src = 0; src < vertices; src++)
      sparse_signal(src, /*current_send_partition=*/0);
    size_t stride = sizeofM<ValueType>(feature_size);
    for (VertexId k = 0; k < rec_count; k++) {
      char *rec = rec_bytes.data() + (size_t)k * stride;
      VertexId vid;
      memcpy(&vid, rec, sizeof(VertexId));
      /* the real driver slots only records whose vertex is a source of an
       * owned edge (graph.hpp:2907-2911) */
      if (subgraphs.empty() || subgraphs[0]->src_get_active(vid))
        sparse_slot(vid, (ValueType *)(rec + sizeof(VertexId)), /*recv_id=*/0);
    }
  }

  /* Single-node per-vertex driver (local_vertex_operation,
   * core/graph.hpp:2351): every vertex against the one local chunk. */
  template <typename R, typename M, typename FN>
  void local_vertex_operation(FN fn, std::vector<CSC_segment_pinned *> &subgraphs,
                              int feature_size, VertexSubset *active) {
    (void)feature_size;
    (void)active;
    for (VertexId vtx = 0; vtx < gnnctx->l_v_num; vtx++)
      fn(vtx, subgraphs[0], /*recv_id=*/0);
  }
};

inline void StubComm::emit_buffer(VertexId vid, ValueType *data, int f_size) {
  g->append_record(vid, data, f_size);
}

#endif /* NTS_REF_STUB_GRAPH_HPP */
