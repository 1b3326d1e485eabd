"""Ring mirror-vertex exchange over torch.distributed (RCCL on GPU, gloo in
CPU tests) for the partitioned aggregation path.

MI355X-native replacement of the reference's MPI host-bounce ring
(/root/reference/comm/network.cpp:524-767 driven by
Graph::sync_compute_decoupled, core/graph.hpp:3640-3719, and
Graph::compute_sync_decoupled, graph.hpp:3456-3622):

  forward  (master -> mirror): each rank owns masters [offs[r], offs[r+1]).
    P-1 ring steps of dense fp32 feature blocks GPU-to-GPU
    (ncclSend/ncclRecv via dist.batch_isend_irecv — no [vid|floats] record
    packing, no pinned-host bounce: mirror ranges are the static partition
    ranges, so the payload is the owner's whole dense block, which is
    exactly what the reference's non-lock-free emit_buffer loop sends,
    network.cpp:476-495).  Each received block feeds the matching chunk's
    CSC aggregation; step s+1's exchange is posted before step s's
    aggregation runs, overlapping transfer with compute (the reference's
    PROC_OVERLAP behavior, graph.hpp:3490-3535, default here).

  backward (mirror -> master): for each remote partition k, the local CSR
    chunk produces partial grads for k's masters; ring-send the dense
    partial block to its owner, who accumulates (the reference serializes
    [vid|floats] records and merges with aggregate_data_buffer_debug,
    cuda/ntsCUDATransferKernel.cuh:49-68 — here the merge is a dense add).

The aggregation itself is injected as `engine` (ops.HipEngine on GPU;
tests inject an oracle-backed engine to run this file's logic on CPU under
gloo, world_size 2 — the product never uses a CPU engine).
"""
import numpy as np
import torch
import torch.distributed as dist


class RingGraph:
    """Rank-local view of the partitioned graph: this rank's chunks (one per
    source partition) plus the global partition offsets."""

    def __init__(self, offs: np.ndarray, rank: int, chunks, device):
        self.offs = [int(o) for o in offs]
        self.rank = rank
        self.world = len(offs) - 1
        self.chunks = chunks          # list of DeviceChunk-like, len == world
        self.device = device
        self.owned_n = self.offs[rank + 1] - self.offs[rank]

    def part_n(self, k):
        return self.offs[k + 1] - self.offs[k]


def _exchange(send_t, dst, recv_t, src):
    """One ring step: send our block to `dst`, receive `src`'s block."""
    ops = [dist.P2POp(dist.isend, send_t, dst),
           dist.P2POp(dist.irecv, recv_t, src)]
    return dist.batch_isend_irecv(ops)


def ring_forward(rg: RingGraph, x_owned: torch.Tensor, engine) -> torch.Tensor:
    """Distributed forward aggregation: returns y over the owned dst range.
    Mirrors sync_compute_decoupled's ring loop (graph.hpp:3685-3707)."""
    P, r = rg.world, rg.rank
    f = x_owned.shape[1]
    assert x_owned.shape[0] == rg.owned_n and x_owned.is_contiguous()
    y = torch.zeros(rg.owned_n, f, dtype=torch.float32, device=x_owned.device)
    if P == 1:
        engine.csc_forward(rg.chunks[r], x_owned, y)
        return y
    maxn = max(rg.part_n(k) for k in range(P))
    bufs = [torch.empty(maxn, f, dtype=torch.float32, device=x_owned.device)
            for _ in range(2)]
    # step 1 posted before local compute (overlap)
    src0 = (r + 1) % P
    reqs = _exchange(x_owned, (r - 1) % P, bufs[0][: rg.part_n(src0)], src0)
    engine.csc_forward(rg.chunks[r], x_owned, y)   # local chunk
    for step in range(1, P):
        src = (r + step) % P
        for rq in reqs:
            rq.wait()
        blk = bufs[(step - 1) % 2][: rg.part_n(src)]
        if step + 1 < P:
            nxt = (r + step + 1) % P
            reqs = _exchange(x_owned, (r - step - 1) % P,
                             bufs[step % 2][: rg.part_n(nxt)], nxt)
        engine.csc_forward(rg.chunks[src], blk, y)
    return y


def ring_backward(rg: RingGraph, grad_y: torch.Tensor, engine) -> torch.Tensor:
    """Distributed backward: returns grad over the owned src range.
    Mirrors compute_sync_decoupled (graph.hpp:3456-3622): local partials per
    remote partition, ring-sent to their owner and accumulated."""
    P, r = rg.world, rg.rank
    f = grad_y.shape[1]
    assert grad_y.shape[0] == rg.owned_n and grad_y.is_contiguous()
    gx = torch.zeros(rg.owned_n, f, dtype=torch.float32, device=grad_y.device)
    engine.csr_backward(rg.chunks[r], grad_y, gx)  # local chunk
    if P == 1:
        return gx
    # pipelined: compute step s+1's partial while step s's exchange is in
    # flight (double-buffered receives; all ranks post exchanges in the same
    # step order, so the grouped sends/recvs match)
    recvs = [torch.empty(rg.owned_n, f, dtype=torch.float32,
                         device=grad_y.device) for _ in range(2)]
    prev = None
    for step in range(1, P):
        k = (r + step) % P          # partition whose masters we feed
        peer_src = (r - step) % P   # rank whose partial for US arrives
        partial = torch.zeros(rg.part_n(k), f, dtype=torch.float32,
                              device=grad_y.device)
        engine.csr_backward(rg.chunks[k], grad_y, partial)
        reqs = _exchange(partial, k, recvs[step % 2], peer_src)
        if prev is not None:
            for rq in prev:
                rq.wait()
            gx += recvs[(step - 1) % 2]
        prev = reqs
    for rq in prev:
        rq.wait()
    gx += recvs[(P - 1) % 2]
    return gx


class DistGPUFuseOp:
    """ForwardGPUfuseOp equivalent (core/ntsDistGPUFusedGraphOp.hpp:48-91):
    forward = ring master->mirror pull, backward = ring mirror->master push.
    Unlike the reference there is no f_input.cpu() host bounce (:58): tensors
    stay in HBM end to end."""

    def __init__(self, rg: RingGraph, engine):
        self.rg = rg
        self.engine = engine

    def forward(self, x_owned):
        return ring_forward(self.rg, x_owned, self.engine)

    def backward(self, grad_y):
        return ring_backward(self.rg, grad_y, self.engine)
