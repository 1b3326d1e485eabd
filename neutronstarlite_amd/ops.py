"""Operator layer mirroring the reference's fused-op surface on torch tensors.

Python mirror of nts::op::ntsGraphOp's calling convention
(/root/reference/core/ntsBaseOp.hpp:24-48): ops take and return new tensors;
`forward` consumes the layer input, `backward` the output grad.  The C++
header mirror for linking toolkits unchanged lives in include/ and
INTEGRATION.md; this module is the host driver used by tests and bench.

  SingleGPUFuseOp  ~ ForwardSingleGPUfuseOp
                     (core/ntsSingleGPUFusedGraphOp.hpp:48-71, driving
                      Graph::forward_single/backward_single, graph.hpp:3806-3855)
  DistGPUFuseOp    ~ ForwardGPUfuseOp
                     (core/ntsDistGPUFusedGraphOp.hpp:48-91, driving
                      sync_compute_decoupled / compute_sync_decoupled) — the
                      ring exchange lives in ring.py on RCCL over xGMI.

All compute runs through the C-ABI HIP shim (shim.Stream) on the caller's
torch stream; there is NO CPU fallback — shim raises if the extension is
missing.
"""
import numpy as np
import torch

from . import shim
from .graph import Chunk


def _u32_cuda(a: np.ndarray, device) -> torch.Tensor:
    """Upload a u32 numpy array as an int32 cuda tensor (same bit pattern)."""
    assert a.dtype == np.uint32
    return torch.from_numpy(a.view(np.int32)).to(device)


class DeviceChunk:
    """A Chunk's CSC+CSR+weights resident in HBM
    (CSC_segment_pinned::CopyGraphToDevice equivalent,
    /root/reference/core/GraphSegment.cpp:178-220)."""

    def __init__(self, ch: Chunk, device):
        self.src_s, self.src_e = ch.src_s, ch.src_e
        self.dst_s, self.dst_e = ch.dst_s, ch.dst_e
        self.edge_size = ch.edge_size
        self.column_offset = _u32_cuda(ch.column_offset, device)
        self.row_indices = _u32_cuda(ch.row_indices, device)
        self.w_fwd = torch.from_numpy(ch.edge_weight_forward).to(device)
        self.row_offset = _u32_cuda(ch.row_offset, device)
        self.column_indices = _u32_cuda(ch.column_indices, device)
        self.w_bwd = torch.from_numpy(ch.edge_weight_backward).to(device)

    @property
    def dst_n(self):
        return self.dst_e - self.dst_s

    @property
    def src_n(self):
        return self.src_e - self.src_s


class HipEngine:
    """The product aggregation engine: HIP kernels via the C-ABI on the
    current torch stream."""

    def __init__(self):
        self.stream = shim.Stream.wrap_torch_current()

    def csc_forward(self, ch: DeviceChunk, x_block: torch.Tensor,
                    y: torch.Tensor, with_weight=True):
        """y[0:dst_n] += CSC-aggregate of x_block (dense over ch's src range)."""
        f = y.shape[1]
        assert x_block.is_cuda and y.is_cuda and x_block.dtype == torch.float32
        assert x_block.is_contiguous() and y.is_contiguous()
        assert x_block.shape[0] == ch.src_n and y.shape[0] == ch.dst_n
        self.stream.gather_by_dst_from_src(
            x_block.data_ptr(), y.data_ptr(), ch.w_fwd.data_ptr(),
            ch.row_indices.data_ptr(), ch.column_offset.data_ptr(),
            ch.src_s, ch.src_e, ch.dst_s, ch.dst_e,
            ch.edge_size, ch.dst_n, f, with_weight)

    def csr_backward(self, ch: DeviceChunk, grad_block: torch.Tensor,
                     out: torch.Tensor, with_weight=True):
        """out[0:src_n] += CSR-aggregate of grad_block (dense over dst range)."""
        f = out.shape[1]
        assert grad_block.is_contiguous() and out.is_contiguous()
        assert grad_block.shape[0] == ch.dst_n and out.shape[0] == ch.src_n
        self.stream.gather_by_src_from_dst(
            grad_block.data_ptr(), out.data_ptr(), ch.w_bwd.data_ptr(),
            ch.row_offset.data_ptr(), ch.column_indices.data_ptr(),
            ch.src_s, ch.src_e, ch.dst_s, ch.dst_e,
            ch.edge_size, ch.src_n, f, with_weight)


class _AggregateFn(torch.autograd.Function):
    """torch.autograd bridge over the fused aggregation: forward = CSC pull,
    backward = CSR push (the ntsGraphOp tape contract of
    ntsContext::self_backward, ntsContext.hpp:276-359, expressed as a torch
    Function so whole training loops are plain torch code)."""

    @staticmethod
    def forward(ctx, x, dchunk, engine):
        ctx.dchunk = dchunk
        ctx.engine = engine
        y = torch.zeros(dchunk.dst_n, x.shape[1], dtype=torch.float32,
                        device=x.device)
        engine.csc_forward(dchunk, x.contiguous(), y)
        return y

    @staticmethod
    def backward(ctx, grad_y):
        gx = torch.zeros(ctx.dchunk.src_n, grad_y.shape[1],
                         dtype=torch.float32, device=grad_y.device)
        ctx.engine.csr_backward(ctx.dchunk, grad_y.contiguous(), gx)
        return gx, None, None


def aggregate(x: torch.Tensor, dchunk: "DeviceChunk",
              engine: "HipEngine") -> torch.Tensor:
    """Autograd-aware fused aggregation (single-GPU chunk)."""
    return _AggregateFn.apply(x, dchunk, engine)


class _MiniBatchAggFn(torch.autograd.Function):
    """torch.autograd bridge over one sampled layer's MiniBatchFuseOp, so
    whole mini-batch training loops (the reference's GCN_CPU_SAMPLE per-layer
    chain, toolkits/GCN_CPU_SAMPLE.hpp:214 + ntsMiniBatchGraphOp.hpp:61-131)
    are plain torch code."""

    @staticmethod
    def forward(ctx, x, op):
        ctx.op = op
        return op.forward(x.contiguous())

    @staticmethod
    def backward(ctx, grad_y):
        return ctx.op.backward(grad_y.contiguous()), None


def minibatch_aggregate(x: torch.Tensor, op: "MiniBatchFuseOp") -> torch.Tensor:
    """Autograd-aware sampled-subgraph aggregation."""
    return _MiniBatchAggFn.apply(x, op)


class MiniBatchFuseOp:
    """MiniBatchFuseOp equivalent (core/ntsMiniBatchGraphOp.hpp:61-131):
    the same aggregation arithmetic on one sampled layer's compacted
    subgraph — forward pulls compacted source rows into destination rows,
    backward pushes destination grads back to compacted sources.  Runs the
    same gfx950 gather kernels with local ids (src range [0, n_src))."""

    def __init__(self, layer, device, engine: "HipEngine" = None):
        self.layer = layer
        self.engine = engine or HipEngine()

        def up(a):
            if isinstance(a, torch.Tensor):       # GPU-resident sampler
                return a.to(device)
            if a.dtype == np.uint32:
                return _u32_cuda(a, device)
            return torch.from_numpy(a).to(device)

        self.column_offset = up(layer.column_offset)
        self.row_indices = up(layer.row_indices_local)
        self.w_fwd = up(layer.edge_weight)
        self.row_offset = up(layer.row_offset)
        self.column_indices = up(layer.column_indices_local)
        self.w_bwd = up(layer.edge_weight_backward)

    def forward(self, x_compact: torch.Tensor) -> torch.Tensor:
        ly = self.layer
        assert x_compact.shape[0] == ly.n_src and x_compact.is_contiguous()
        f = x_compact.shape[1]
        y = torch.zeros(ly.n_dst, f, dtype=torch.float32,
                        device=x_compact.device)
        self.engine.stream.gather_by_dst_from_src(
            x_compact.data_ptr(), y.data_ptr(), self.w_fwd.data_ptr(),
            self.row_indices.data_ptr(), self.column_offset.data_ptr(),
            0, ly.n_src, 0, ly.n_dst, ly.e_size, ly.n_dst, f, True)
        return y

    def backward(self, grad_y: torch.Tensor) -> torch.Tensor:
        ly = self.layer
        assert grad_y.shape[0] == ly.n_dst and grad_y.is_contiguous()
        f = grad_y.shape[1]
        gx = torch.zeros(ly.n_src, f, dtype=torch.float32,
                         device=grad_y.device)
        self.engine.stream.gather_by_src_from_dst(
            grad_y.data_ptr(), gx.data_ptr(), self.w_bwd.data_ptr(),
            self.row_offset.data_ptr(), self.column_indices.data_ptr(),
            0, ly.n_src, 0, ly.n_dst, ly.e_size, ly.n_src, f, True)
        return gx


class SingleGPUFuseOp:
    """ForwardSingleGPUfuseOp equivalent: whole graph as one chunk on one GPU
    (core/ntsSingleGPUFusedGraphOp.hpp:48-71)."""

    def __init__(self, dchunk: DeviceChunk, engine: HipEngine = None):
        self.ch = dchunk
        self.engine = engine or HipEngine()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = torch.zeros(self.ch.dst_n, x.shape[1], dtype=torch.float32,
                        device=x.device)
        self.engine.csc_forward(self.ch, x, y)
        return y

    def backward(self, grad_y: torch.Tensor) -> torch.Tensor:
        gx = torch.zeros(self.ch.src_n, grad_y.shape[1], dtype=torch.float32,
                         device=grad_y.device)
        self.engine.csr_backward(self.ch, grad_y, gx)
        return gx
