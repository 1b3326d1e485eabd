/* nts_comm.cpp — RCCL implementation of include/nts_comm.h.
 *
 * Replaces the reference's MPI ring communicator
 * (/root/reference/comm/network.cpp:524-767) with stream-ordered RCCL
 * grouped p2p over xGMI; see the header for the entry-point map.
 *
 * RCCL is resolved at RUNTIME via dlopen/dlsym, not DT_NEEDED: the torch
 * wheel bundles its own librccl (SONAME librccl.so.1, FILE name librccl.so)
 * and /opt/rocm ships another; linking either statically put BOTH copies in
 * one process (first-load interposition over two half-initialized runtimes
 * -> heap corruption, observed).  dlopen with RTLD_NOLOAD first reuses
 * whichever copy the process already loaded (torch's, in any torch-linked
 * binary or python process), falling back to loading one fresh.
 */
#include <dlfcn.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>

#include <rccl/rccl.h> /* types/enums only; functions come from dlsym */

#include "nts_comm.h"

#define DIE(...)                              \
  do {                                        \
    std::fprintf(stderr, "nts_comm: " __VA_ARGS__); \
    std::fprintf(stderr, "\n");               \
    std::abort(); /* reference CHECK culture, ntsCUDAGraphOP.cu:13-19 */ \
  } while (0)

namespace {

struct RcclApi {
  ncclResult_t (*CommInitAll)(ncclComm_t *, int, const int *);
  ncclResult_t (*GetUniqueId)(ncclUniqueId *);
  ncclResult_t (*CommInitRank)(ncclComm_t *, int, ncclUniqueId, int);
  ncclResult_t (*CommDestroy)(ncclComm_t);
  ncclResult_t (*CommUserRank)(const ncclComm_t, int *);
  ncclResult_t (*CommCount)(const ncclComm_t, int *);
  ncclResult_t (*GroupStart)(void);
  ncclResult_t (*GroupEnd)(void);
  ncclResult_t (*Send)(const void *, size_t, ncclDataType_t, int, ncclComm_t,
                       hipStream_t);
  ncclResult_t (*Recv)(void *, size_t, ncclDataType_t, int, ncclComm_t,
                       hipStream_t);
  ncclResult_t (*AllReduce)(const void *, void *, size_t, ncclDataType_t,
                            ncclRedOp_t, ncclComm_t, hipStream_t);
  ncclResult_t (*Broadcast)(const void *, void *, size_t, ncclDataType_t, int,
                            ncclComm_t, hipStream_t);
  const char *(*GetErrorString)(ncclResult_t);
};

RcclApi *api() {
  static RcclApi *a = [] {
    void *h = dlopen("librccl.so", RTLD_NOW | RTLD_NOLOAD | RTLD_GLOBAL);
    if (!h) h = dlopen("librccl.so.1", RTLD_NOW | RTLD_NOLOAD | RTLD_GLOBAL);
    if (!h) h = dlopen("librccl.so", RTLD_NOW | RTLD_GLOBAL);
    if (!h) h = dlopen("librccl.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) DIE("no librccl found (dlopen: %s)", dlerror());
    auto *a = new RcclApi;
    auto sym = [&](const char *name) {
      void *p = dlsym(h, name);
      if (!p) DIE("librccl lacks %s", name);
      return p;
    };
    a->CommInitAll = (decltype(a->CommInitAll))sym("ncclCommInitAll");
    a->GetUniqueId = (decltype(a->GetUniqueId))sym("ncclGetUniqueId");
    a->CommInitRank = (decltype(a->CommInitRank))sym("ncclCommInitRank");
    a->CommDestroy = (decltype(a->CommDestroy))sym("ncclCommDestroy");
    a->CommUserRank = (decltype(a->CommUserRank))sym("ncclCommUserRank");
    a->CommCount = (decltype(a->CommCount))sym("ncclCommCount");
    a->GroupStart = (decltype(a->GroupStart))sym("ncclGroupStart");
    a->GroupEnd = (decltype(a->GroupEnd))sym("ncclGroupEnd");
    a->Send = (decltype(a->Send))sym("ncclSend");
    a->Recv = (decltype(a->Recv))sym("ncclRecv");
    a->AllReduce = (decltype(a->AllReduce))sym("ncclAllReduce");
    a->Broadcast = (decltype(a->Broadcast))sym("ncclBroadcast");
    a->GetErrorString = (decltype(a->GetErrorString))sym("ncclGetErrorString");
    return a;
  }();
  return a;
}

void check(ncclResult_t r, const char *what) {
  if (r != ncclSuccess)
    DIE("%s failed: %s", what, api()->GetErrorString(r));
}

hipStream_t raw(nts_stream *s) { return (hipStream_t)nts_stream_handle(s); }

}  // namespace

struct nts_comm {
  ncclComm_t comm;
};

static_assert(NTS_COMM_UNIQUE_ID_BYTES == NCCL_UNIQUE_ID_BYTES,
              "unique-id size drifted from RCCL's");

extern "C" int nts_comm_init_all(nts_comm **comms, int ndev,
                                 const int *devices) {
  ncclComm_t *cs = (ncclComm_t *)std::malloc(sizeof(ncclComm_t) * ndev);
  check(api()->CommInitAll(cs, ndev, devices), "ncclCommInitAll");
  for (int i = 0; i < ndev; i++) comms[i] = new nts_comm{cs[i]};
  std::free(cs);
  return 0;
}

extern "C" void nts_comm_unique_id(char uid[NTS_COMM_UNIQUE_ID_BYTES]) {
  ncclUniqueId id;
  check(api()->GetUniqueId(&id), "ncclGetUniqueId");
  std::memcpy(uid, id.internal, NTS_COMM_UNIQUE_ID_BYTES);
}

extern "C" int nts_comm_init_rank(nts_comm **comm, int nranks,
                                  const char uid[NTS_COMM_UNIQUE_ID_BYTES],
                                  int rank) {
  ncclUniqueId id;
  std::memcpy(id.internal, uid, NTS_COMM_UNIQUE_ID_BYTES);
  ncclComm_t c;
  check(api()->CommInitRank(&c, nranks, id, rank), "ncclCommInitRank");
  *comm = new nts_comm{c};
  return 0;
}

extern "C" void nts_comm_destroy(nts_comm *c) {
  if (!c) return;
  api()->CommDestroy(c->comm);
  delete c;
}

extern "C" int nts_comm_rank(nts_comm *c) {
  int r;
  check(api()->CommUserRank(c->comm, &r), "ncclCommUserRank");
  return r;
}

extern "C" int nts_comm_size(nts_comm *c) {
  int n;
  check(api()->CommCount(c->comm, &n), "ncclCommCount");
  return n;
}

extern "C" void nts_comm_group_begin(void) {
  check(api()->GroupStart(), "ncclGroupStart");
}
extern "C" void nts_comm_group_end(void) {
  check(api()->GroupEnd(), "ncclGroupEnd");
}

extern "C" void nts_comm_send_f32(nts_comm *c, nts_stream *s, const float *buf,
                                  long n, int peer) {
  check(api()->Send(buf, (size_t)n, ncclFloat32, peer, c->comm, raw(s)),
        "ncclSend");
}

extern "C" void nts_comm_recv_f32(nts_comm *c, nts_stream *s, float *buf,
                                  long n, int peer) {
  check(api()->Recv(buf, (size_t)n, ncclFloat32, peer, c->comm, raw(s)),
        "ncclRecv");
}

extern "C" void nts_comm_allreduce_sum_f32(nts_comm *c, nts_stream *s,
                                           const float *in, float *out,
                                           long n) {
  check(api()->AllReduce(in, out, (size_t)n, ncclFloat32, ncclSum, c->comm,
                         raw(s)),
        "ncclAllReduce");
}

extern "C" void nts_comm_bcast_f32(nts_comm *c, nts_stream *s, float *buf,
                                   long n, int root) {
  check(api()->Broadcast(buf, buf, (size_t)n, ncclFloat32, root, c->comm,
                         raw(s)),
        "ncclBroadcast");
}
