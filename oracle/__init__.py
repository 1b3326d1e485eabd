"""ctypes wrapper for the CPU parity oracle.

TEST INFRASTRUCTURE ONLY: importable by tests/, __graft_entry__.smoke() and
bench.py's cpu_baseline leg — never by the product path (see oracle.c header).
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "liboracle.so")

_lib = None


def build():
    subprocess.run(["make", "-C", _DIR], check=True, capture_output=True)


def _load():
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_SO):
        build()
    lib = ctypes.CDLL(_SO)
    u32p = ctypes.POINTER(ctypes.c_uint32)
    f32p = ctypes.POINTER(ctypes.c_float)
    lib.oracle_degrees.argtypes = [u32p, ctypes.c_uint64, ctypes.c_uint32, u32p, u32p]
    lib.oracle_norm_weights.argtypes = [u32p, u32p, ctypes.c_uint64, u32p, u32p, f32p]
    lib.oracle_csc_forward.argtypes = [u32p, u32p, f32p, f32p, f32p,
                                       ctypes.c_uint32, ctypes.c_uint32, ctypes.c_int64]
    lib.oracle_csr_backward.argtypes = [u32p, u32p, f32p, f32p, f32p,
                                        ctypes.c_uint32, ctypes.c_uint32, ctypes.c_int64]
    lib.oracle_deserialize.argtypes = [f32p, ctypes.c_uint64, ctypes.c_uint32, f32p, ctypes.c_int64]
    lib.oracle_agg_msg_to_master.argtypes = [f32p, f32p, ctypes.c_uint64, ctypes.c_uint32, ctypes.c_int64]
    lib.oracle_num_threads.restype = ctypes.c_int
    lib.oracle_scatter_src_to_msg.argtypes = [f32p, f32p, u32p, u32p, u32p,
                                              ctypes.c_uint32, ctypes.c_int64]
    lib.oracle_gather_msg_to_src.argtypes = [f32p, f32p, u32p, u32p, u32p,
                                             ctypes.c_uint32, ctypes.c_int64]
    lib.oracle_scatter_dst_to_msg.argtypes = [f32p, f32p, u32p,
                                              ctypes.c_uint32, ctypes.c_int64]
    lib.oracle_gather_msg_to_dst.argtypes = [f32p, f32p, u32p,
                                             ctypes.c_uint32, ctypes.c_int64]
    lib.oracle_scatter_grad_back_to_msg.argtypes = [f32p, f32p, u32p,
                                                    ctypes.c_uint32, ctypes.c_int64]
    lib.oracle_edge_softmax_forward.argtypes = [f32p, f32p, f32p, u32p,
                                                ctypes.c_uint32, ctypes.c_int64]
    lib.oracle_edge_softmax_backward.argtypes = [f32p, f32p, f32p, u32p,
                                                 ctypes.c_uint32, ctypes.c_int64]
    _lib = lib
    return lib


def _u32(a):
    assert a.dtype == np.uint32 and a.flags.c_contiguous
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32))


def _f32(a):
    assert a.dtype == np.float32 and a.flags.c_contiguous
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_float))


def degrees(edges, v):
    """edges: (E,2) u32 (src,dst). Returns (out_degree, in_degree), clamped >=1."""
    lib = _load()
    edges = np.ascontiguousarray(edges, dtype=np.uint32)
    outd = np.zeros(v, dtype=np.uint32)
    ind = np.zeros(v, dtype=np.uint32)
    lib.oracle_degrees(_u32(edges.reshape(-1)), edges.shape[0], v, _u32(outd), _u32(ind))
    return outd, ind


def norm_weights(src, dst, out_degree, in_degree):
    lib = _load()
    w = np.empty(len(src), dtype=np.float32)
    lib.oracle_norm_weights(_u32(src), _u32(dst), len(src),
                            _u32(out_degree), _u32(in_degree), _f32(w))
    return w


def csc_forward(column_offset, row_indices, weight, x, src_s, dst_n, f, out=None):
    lib = _load()
    if out is None:
        out = np.zeros((dst_n, f), dtype=np.float32)
    lib.oracle_csc_forward(_u32(column_offset), _u32(row_indices), _f32(weight),
                           _f32(x), _f32(out), src_s, dst_n, f)
    return out


def csr_backward(row_offset, column_indices, weight, grad, dst_s, src_n, f, out=None):
    lib = _load()
    if out is None:
        out = np.zeros((src_n, f), dtype=np.float32)
    lib.oracle_csr_backward(_u32(row_offset), _u32(column_indices), _f32(weight),
                            _f32(grad), _f32(out), dst_s, src_n, f)
    return out


def deserialize(msg, count, part_start, dense_rows, f):
    lib = _load()
    lib.oracle_deserialize(_f32(msg), count, part_start, _f32(dense_rows), f)
    return dense_rows


def agg_msg_to_master(master, msg, count, part_start, f):
    lib = _load()
    lib.oracle_agg_msg_to_master(_f32(master), _f32(msg), count, part_start, f)
    return master


def num_threads():
    return _load().oracle_num_threads()


def scatter_src_to_msg(msg, mirror, row_indices, column_offset, mirror_index,
                       batch, f):
    _load().oracle_scatter_src_to_msg(_f32(msg), _f32(mirror),
                                      _u32(row_indices), _u32(column_offset),
                                      _u32(mirror_index), batch, f)
    return msg


def gather_msg_to_src(mirror, msg, row_indices, column_offset, mirror_index,
                      batch, f):
    _load().oracle_gather_msg_to_src(_f32(mirror), _f32(msg),
                                     _u32(row_indices), _u32(column_offset),
                                     _u32(mirror_index), batch, f)
    return mirror


def scatter_dst_to_msg(msg, dst_feat, column_offset, batch, f):
    _load().oracle_scatter_dst_to_msg(_f32(msg), _f32(dst_feat),
                                      _u32(column_offset), batch, f)
    return msg


def gather_msg_to_dst(dst_feat, msg, column_offset, batch, f):
    _load().oracle_gather_msg_to_dst(_f32(dst_feat), _f32(msg),
                                     _u32(column_offset), batch, f)
    return dst_feat


def scatter_grad_back_to_msg(input_grad, msg_grad, column_offset, batch, f):
    _load().oracle_scatter_grad_back_to_msg(_f32(input_grad), _f32(msg_grad),
                                            _u32(column_offset), batch, f)
    return msg_grad


def edge_softmax_forward(out, inp, cached, column_offset, batch, f):
    _load().oracle_edge_softmax_forward(_f32(out), _f32(inp), _f32(cached),
                                        _u32(column_offset), batch, f)
    return out


def edge_softmax_backward(in_grad, out_grad, cached, column_offset, batch, f):
    _load().oracle_edge_softmax_backward(_f32(in_grad), _f32(out_grad),
                                         _f32(cached), _u32(column_offset),
                                         batch, f)
    return in_grad
