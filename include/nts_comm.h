/* nts_comm.h — C-ABI of the RCCL ring communicator (additive to nts_hip.h).
 *
 * MI355X-native replacement of the reference's MPI communicator for the C++
 * operator layer: NtsGraphCommunicator's P-step ring of MPI_Send/MPI_Probe/
 * MPI_Recv host-bounce threads (/root/reference/comm/network.cpp:524-767)
 * becomes grouped ncclSend/ncclRecv of dense fp32 row blocks GPU-to-GPU over
 * xGMI, ordered on a HIP stream — no pinned-host staging, no MPI on the GPU
 * path (north_star clause).  The one-process-per-GPU Python path uses
 * torch.distributed (RCCL) directly and never loads this library; this ABI
 * serves the C++ ForwardGPUfuseOp (cpp/include/nts/nts.hpp) and covers both
 * process models:
 *   - single process driving N devices: nts_comm_init_all (ncclCommInitAll)
 *   - one process per rank: nts_comm_unique_id + nts_comm_init_rank
 *     (ncclGetUniqueId / ncclCommInitRank; the 128-byte id travels by the
 *     caller's own side channel, standing in for MPI_Bcast of the id)
 *
 * Entry-point map (reference -> here):
 *   network.cpp:524-767 ring send/recv threads  -> nts_comm_group_begin/end
 *        + nts_comm_send_f32/nts_comm_recv_f32 on the comm stream
 *   network.h:198-203 Network_simple::all_reduce_sum (weight grads)
 *        -> nts_comm_allreduce_sum_f32
 *   network.h:208-211 MPI_Bcast of initial weights -> nts_comm_bcast_f32
 *
 * Error culture: abort on any nccl/hip error (reference CHECK semantics,
 * cuda/ntsCUDAGraphOP.cu:13-19).  All calls are stream-ordered on the
 * nts_stream passed; host does not block unless stated.
 */
#ifndef NTS_COMM_H
#define NTS_COMM_H

#include "nts_hip.h"

#ifdef __cplusplus
extern "C" {
#endif

typedef struct nts_comm nts_comm;

#define NTS_COMM_UNIQUE_ID_BYTES 128 /* == NCCL_UNIQUE_ID_BYTES */

/* Single-process multi-GPU: create one communicator per device.
 * comms[i] talks for devices[i] (NULL devices -> 0..ndev-1). */
int nts_comm_init_all(nts_comm **comms, int ndev, const int *devices);

/* One process per rank. */
void nts_comm_unique_id(char uid[NTS_COMM_UNIQUE_ID_BYTES]);
int nts_comm_init_rank(nts_comm **comm, int nranks,
                       const char uid[NTS_COMM_UNIQUE_ID_BYTES], int rank);

void nts_comm_destroy(nts_comm *c);
int nts_comm_rank(nts_comm *c);
int nts_comm_size(nts_comm *c);

/* Group p2p calls between begin/end so RCCL launches them as one fused
 * kernel set (the ring step: each rank's send+recv posted together). */
void nts_comm_group_begin(void);
void nts_comm_group_end(void);

void nts_comm_send_f32(nts_comm *c, nts_stream *s, const float *buf,
                       long n, int peer);
void nts_comm_recv_f32(nts_comm *c, nts_stream *s, float *buf,
                       long n, int peer);

void nts_comm_allreduce_sum_f32(nts_comm *c, nts_stream *s, const float *in,
                                float *out, long n); /* in==out: in place */
void nts_comm_bcast_f32(nts_comm *c, nts_stream *s, float *buf, long n,
                        int root);

#ifdef __cplusplus
}
#endif

#endif /* NTS_COMM_H */
