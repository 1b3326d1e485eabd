"""ctypes wrapper for oracle/_ref/libntsref.so — REFERENCE-EXECUTED code.

TEST INFRASTRUCTURE ONLY (same rules as oracle/__init__.py): the library is
the reference's own hot-path sources compiled from /root/reference by
oracle/ref_harness/ (nothing copied; see the Makefile there).  It anchors
parity as `kind: "reference"`: golden fixtures are generated
from it and oracle/oracle.c must match it bit-exactly.

Built in the dev container (make -C oracle/ref_harness); the prebuilt .so
travels to GPU boxes via the snapshot.  `available()` is False when neither
the .so nor /root/reference exists.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "_ref", "libntsref.so")

_lib = None


def build():
    subprocess.run(["make", "-C", os.path.join(_DIR, "ref_harness")],
                   check=True, capture_output=True)


def available():
    if os.path.exists(_SO):
        return True
    if os.path.isdir("/root/reference"):
        try:
            build()
        except subprocess.CalledProcessError:
            return False
        return os.path.exists(_SO)
    return False


def _load():
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_SO) and os.path.isdir("/root/reference"):
        build()
    lib = ctypes.CDLL(_SO)
    u32p = ctypes.POINTER(ctypes.c_uint32)
    f32p = ctypes.POINTER(ctypes.c_float)
    u32, u64, i64 = ctypes.c_uint32, ctypes.c_uint64, ctypes.c_int64
    lib.nts_ref_comp.argtypes = [f32p, f32p, ctypes.c_float, ctypes.c_int]
    lib.nts_ref_acc.argtypes = [f32p, f32p, ctypes.c_int]
    lib.nts_ref_norm_degree.argtypes = [u32, u32, u32p, u32p, u32]
    lib.nts_ref_norm_degree.restype = ctypes.c_float
    for fn in (lib.nts_ref_fused_forward, lib.nts_ref_fused_backward):
        fn.argtypes = [u32, i64, u32p, u32p, u32p, u32p, u32p, u32p, f32p, f32p]
    for fn in (lib.nts_ref_src_scatter_fwd, lib.nts_ref_src_scatter_bwd,
               lib.nts_ref_dst_aggregate_fwd, lib.nts_ref_dst_aggregate_bwd):
        fn.argtypes = [u32, u64, i64, u32p, u32p, f32p, f32p]
    for fn in (lib.nts_ref_minibatch_forward, lib.nts_ref_minibatch_backward):
        fn.argtypes = [u32, u32, u32, i64, u32p, u32p, u32p, u32p, u32p,
                       u32p, f32p, f32p]
    lib.nts_ref_dist_mirror_index.argtypes = [u32, u32, u32p, u32p, u32p]
    lib.nts_ref_dist_mirror_index.restype = u32
    for fn in (lib.nts_ref_dist_get_dep_nbr_fwd,
               lib.nts_ref_dist_get_dep_nbr_bwd,
               lib.nts_ref_dist_scatter_src_fwd,
               lib.nts_ref_dist_scatter_src_bwd,
               lib.nts_ref_dist_aggregate_dst_fwd,
               lib.nts_ref_dist_aggregate_dst_bwd):
        fn.argtypes = [u32, u32, i64, u32p, u32p, f32p, f32p]
    lib.nts_ref_ok.restype = ctypes.c_int
    assert lib.nts_ref_ok() == 1
    _lib = lib
    return lib


def _u32(a):
    assert a.dtype == np.uint32 and a.flags.c_contiguous
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32))


def _f32(a):
    assert a.dtype == np.float32 and a.flags.c_contiguous
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_float))


def comp(out, inp, weight, f):
    """nts_comp (core/ntsBaseOp.hpp:82-104), reference-compiled."""
    _load().nts_ref_comp(_f32(out), _f32(inp), weight, f)
    return out


def acc(out, inp, f):
    """nts_acc (core/ntsBaseOp.hpp:114-119), reference-compiled."""
    _load().nts_ref_acc(_f32(out), _f32(inp), f)
    return out


def norm_degree(src, dst, out_degree, in_degree):
    """nts_norm_degree (core/ntsBaseOp.hpp:194-197), reference-compiled."""
    return _load().nts_ref_norm_degree(src, dst, _u32(out_degree),
                                       _u32(in_degree), len(out_degree))


def fused_forward(v, f, col_off, rows, row_off, cols, outdeg, indeg, x):
    """ForwardCPUfuseOp::forward at 1 rank (ntsCPUFusedGraphOp.hpp:41-109)."""
    y = np.zeros((v, f), dtype=np.float32)
    _load().nts_ref_fused_forward(v, f, _u32(col_off), _u32(rows),
                                  _u32(row_off), _u32(cols), _u32(outdeg),
                                  _u32(indeg), _f32(x), _f32(y))
    return y


def fused_backward(v, f, col_off, rows, row_off, cols, outdeg, indeg, gy):
    """ForwardCPUfuseOp::backward at 1 rank (ntsCPUFusedGraphOp.hpp:110-167)."""
    gx = np.zeros((v, f), dtype=np.float32)
    _load().nts_ref_fused_backward(v, f, _u32(col_off), _u32(rows),
                                   _u32(row_off), _u32(cols), _u32(outdeg),
                                   _u32(indeg), _f32(gy), _f32(gx))
    return gx


def src_scatter_fwd(v, e, f, col_off, rows, x):
    """SingleCPUSrcScatterOp::forward (ntsSingleCPUGraphOp.hpp:102-122)."""
    msg = np.zeros((e, f), dtype=np.float32)
    _load().nts_ref_src_scatter_fwd(v, e, f, _u32(col_off), _u32(rows),
                                    _f32(x), _f32(msg))
    return msg


def src_scatter_bwd(v, e, f, col_off, rows, msg_grad):
    """SingleCPUSrcScatterOp::backward (ntsSingleCPUGraphOp.hpp:124-145).
    NOTE: the reference swaps nts_acc's arguments here (accumulates the
    zeroed input grad INTO the output grad) and so always returns zeros —
    a reference bug we document and do NOT replicate (our kernels follow the
    GPU twin gather_msg_to_src_mirror, cuda/ntsCUDADistKernel.cuh:46-63)."""
    gx = np.zeros((v, f), dtype=np.float32)
    _load().nts_ref_src_scatter_bwd(v, e, f, _u32(col_off), _u32(rows),
                                    _f32(msg_grad), _f32(gx))
    return gx


def dist_mirror_index(v, col_off, rows):
    """generateMirrorIndex (PartitionedGraph.hpp:295-305), reference
    semantics: prefix sum; returns (MirrorIndex[v+1], owned_mirrors)."""
    e = int(col_off[-1])
    out = np.zeros(v + 1, dtype=np.uint32)
    n = _load().nts_ref_dist_mirror_index(v, e, _u32(col_off), _u32(rows),
                                          _u32(out))
    return out, int(n)


def _dist(name, v, f, col_off, rows, inp, out_rows):
    e = int(col_off[-1])
    out = np.zeros((out_rows, f), dtype=np.float32)
    getattr(_load(), name)(v, e, f, _u32(col_off), _u32(rows), _f32(inp),
                           _f32(out))
    return out


def dist_get_dep_nbr_fwd(v, f, col_off, rows, x, n_mirrors):
    """DistGetDepNbrOp::forward at 1 rank (ntsDistCPUGraphOp.hpp:42-83)."""
    return _dist("nts_ref_dist_get_dep_nbr_fwd", v, f, col_off, rows, x,
                 n_mirrors)


def dist_get_dep_nbr_bwd(v, f, col_off, rows, mirror_grad):
    """DistGetDepNbrOp::backward (ntsDistCPUGraphOp.hpp:85-124)."""
    return _dist("nts_ref_dist_get_dep_nbr_bwd", v, f, col_off, rows,
                 mirror_grad, v)


def dist_scatter_src_fwd(v, f, col_off, rows, mirror):
    """DistScatterSrc::forward (ntsDistCPUGraphOp.hpp:135-159)."""
    e = int(col_off[-1])
    return _dist("nts_ref_dist_scatter_src_fwd", v, f, col_off, rows,
                 mirror, e)


def dist_scatter_src_bwd(v, f, col_off, rows, msg_grad, n_mirrors):
    """DistScatterSrc::backward (ntsDistCPUGraphOp.hpp:161-184)."""
    return _dist("nts_ref_dist_scatter_src_bwd", v, f, col_off, rows,
                 msg_grad, n_mirrors)


def dist_aggregate_dst_fwd(v, f, col_off, rows, msg):
    """DistAggregateDst::forward (ntsDistCPUGraphOp.hpp:251-277)."""
    return _dist("nts_ref_dist_aggregate_dst_fwd", v, f, col_off, rows,
                 msg, v)


def dist_aggregate_dst_bwd(v, f, col_off, rows, y_grad):
    """DistAggregateDst::backward (ntsDistCPUGraphOp.hpp:279-305)."""
    e = int(col_off[-1])
    return _dist("nts_ref_dist_aggregate_dst_bwd", v, f, col_off, rows,
                 y_grad, e)


def dst_aggregate_fwd(v, e, f, col_off, rows, msg):
    """SingleCPUDstAggregateOp::forward (ntsSingleCPUGraphOp.hpp:157-179)."""
    y = np.zeros((v, f), dtype=np.float32)
    _load().nts_ref_dst_aggregate_fwd(v, e, f, _u32(col_off), _u32(rows),
                                      _f32(msg), _f32(y))
    return y


def minibatch_forward(v, f, col_off, r_i_local, dst_ids, src_ids, outdeg,
                      indeg, x):
    """MiniBatchFuseOp::forward (ntsMiniBatchGraphOp.hpp:71-101): sampled
    CSC aggregation with norm-degree weights over GLOBAL degrees."""
    n_dst, n_src = len(dst_ids), len(src_ids)
    y = np.zeros((n_dst, f), dtype=np.float32)
    _load().nts_ref_minibatch_forward(v, n_dst, n_src, f, _u32(col_off),
                                      _u32(r_i_local), _u32(dst_ids),
                                      _u32(src_ids), _u32(outdeg),
                                      _u32(indeg), _f32(x), _f32(y))
    return y


def minibatch_backward(v, f, col_off, r_i_local, dst_ids, src_ids, outdeg,
                       indeg, gy):
    """MiniBatchFuseOp::backward (ntsMiniBatchGraphOp.hpp:102-129)."""
    n_dst, n_src = len(dst_ids), len(src_ids)
    gx = np.zeros((n_src, f), dtype=np.float32)
    _load().nts_ref_minibatch_backward(v, n_dst, n_src, f, _u32(col_off),
                                       _u32(r_i_local), _u32(dst_ids),
                                       _u32(src_ids), _u32(outdeg),
                                       _u32(indeg), _f32(gy), _f32(gx))
    return gx


def dst_aggregate_bwd(v, e, f, col_off, rows, y_grad):
    """SingleCPUDstAggregateOp::backward (ntsSingleCPUGraphOp.hpp:181-203)."""
    mg = np.zeros((e, f), dtype=np.float32)
    _load().nts_ref_dst_aggregate_bwd(v, e, f, _u32(col_off), _u32(rows),
                                      _f32(y_grad), _f32(mg))
    return mg
