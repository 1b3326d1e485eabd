"""ctypes binding of the C-ABI in include/nts_hip.h (the product compute path).

FAILS LOUDLY if the HIP extension is missing: there is no CPU fallback in the
product path — the oracle under oracle/ is test infrastructure only.
"""
import atexit
import ctypes
import os
import weakref

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libnts_hip.so")

# kernel-time tags (enum nts_ktag in include/nts_hip.h)
KTAG_FWD = 0
KTAG_BWD = 1
KTAG_DESER = 2
KTAG_AGGMSG = 3
KTAG_ITEMS = 4
KTAG_EDGE = 5

_c = ctypes
_vp = _c.c_void_p
_u32 = _c.c_uint32
_i32 = _c.c_int
_i64 = _c.c_long


class NtsHipMissing(RuntimeError):
    pass


_lib = None


def lib():
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_SO):
        raise NtsHipMissing(
            f"HIP extension not built: {_SO} missing. Run "
            "`python -m neutronstarlite_amd.build` (or __graft_entry__.build()). "
            "The product path has no CPU fallback.")
    l = ctypes.CDLL(_SO)
    # streams / timing
    l.nts_stream_create.restype = _vp
    l.nts_stream_wrap.restype = _vp
    l.nts_stream_wrap.argtypes = [_vp]
    l.nts_stream_destroy.argtypes = [_vp]
    l.nts_stream_sync.argtypes = [_vp]
    l.nts_stream_handle.restype = _vp
    l.nts_stream_handle.argtypes = [_vp]
    l.nts_stream_timing.argtypes = [_vp, _i32]
    l.nts_stream_timing_reset.argtypes = [_vp]
    l.nts_stream_kernel_ns.restype = _c.c_double
    l.nts_stream_kernel_ns.argtypes = [_vp, _i32]
    l.nts_stream_kernel_launches.restype = _c.c_longlong
    l.nts_stream_kernel_launches.argtypes = [_vp, _i32]
    # memory
    l.nts_malloc_gpu.restype = _vp
    l.nts_malloc_gpu.argtypes = [_i64]
    l.nts_malloc_pinned.restype = _vp
    l.nts_malloc_pinned.argtypes = [_i64]
    l.nts_get_device_pointer.restype = _vp
    l.nts_get_device_pointer.argtypes = [_vp]
    l.nts_free_gpu.argtypes = [_vp]
    l.nts_free_host.argtypes = [_vp]
    l.nts_zero_buffer.argtypes = [_vp, _vp, _i64]
    l.nts_memcpy_h2d.argtypes = [_vp, _vp, _vp, _i64, _i32]
    l.nts_memcpy_d2h.argtypes = [_vp, _vp, _vp, _i64, _i32]
    # hot kernels
    l.nts_gather_by_dst_from_src.argtypes = [_vp] + [_vp] * 5 + [_u32] * 7 + [_i32]
    l.nts_gather_by_src_from_dst.argtypes = [_vp] + [_vp] * 5 + [_u32] * 7 + [_i32]
    l.nts_items_cache_clear.argtypes = [_vp]
    l.nts_deserialize_to_gpu.argtypes = [_vp, _vp, _vp, _u32, _u32, _u32, _u32, _i32]
    l.nts_aggregate_comm_result.argtypes = [_vp, _vp, _vp, _u32, _u32, _u32, _u32, _i32]
    l.nts_gather_rows.argtypes = [_vp, _vp, _vp, _vp, _u32, _u32, _u32]
    l.nts_scatter_rows.argtypes = [_vp, _vp, _vp, _vp, _u32, _u32, _u32]
    l.nts_scatter_add_rows.argtypes = [_vp, _vp, _vp, _vp, _u32, _u32, _u32]
    # edge kernels
    l.nts_scatter_src_mirror_to_msg.argtypes = [_vp] + [_vp] * 5 + [_u32] * 2
    l.nts_gather_msg_to_src_mirror.argtypes = [_vp] + [_vp] * 5 + [_u32] * 2
    l.nts_scatter_dst_to_msg.argtypes = [_vp] + [_vp] * 4 + [_u32] * 2
    l.nts_gather_msg_to_dst.argtypes = [_vp] + [_vp] * 4 + [_u32] * 2
    l.nts_scatter_grad_back_to_message.argtypes = [_vp] + [_vp] * 4 + [_u32] * 2
    l.nts_edge_dot.argtypes = [_vp] + [_vp] * 5 + [_u32] * 3
    l.nts_edge_softmax_forward.argtypes = [_vp] + [_vp] * 5 + [_u32] * 2
    l.nts_edge_softmax_backward.argtypes = [_vp] + [_vp] * 5 + [_u32] * 2
    l.nts_edge_softmax_forward_dual.argtypes = [_vp] + [_vp] * 6 + [_u32] * 2
    l.nts_edge_softmax_backward_fused.argtypes = (
        [_vp] + [_vp] * 6 + [_c.c_float] + [_vp] * 2 + [_u32] * 2)
    l.nts_gather_by_src_from_dst_dot.argtypes = (
        [_vp] + [_vp] * 5 + [_u32] * 4 + [_vp] * 3)
    l.nts_gather_by_src_from_dst_dot.restype = _i32
    l.nts_edge_attention_forward.argtypes = (
        [_vp] + [_vp] * 8 + [_c.c_float] + [_vp] + [_u32])
    l.nts_items_reuse.argtypes = [_vp, _i32]
    l.nts_weight_sum.argtypes = [_vp, _vp, _vp, _vp, _u32]
    l.nts_permute_f32.argtypes = [_vp, _vp, _vp, _vp, _i64]
    l.nts_sample_reservoir.argtypes = [_vp, _vp, _vp, _vp, _u32, _u32,
                                       _c.c_ulonglong, _vp, _vp]
    l.nts_sample_reservoir_dbg_fallback.argtypes = l.nts_sample_reservoir.argtypes
    l.nts_device_count.restype = _i32
    l.nts_set_device.argtypes = [_i32]
    l.nts_build_arch.restype = _c.c_char_p
    _lib = l
    return l


_live_streams = weakref.WeakSet()


@atexit.register
def _destroy_streams_at_exit():
    # Destroy stream objects (draining events, freeing item caches) while the
    # HIP runtime is still alive; leaking them into runtime teardown can
    # abort the process at exit.
    for s in list(_live_streams):
        try:
            s.destroy()
        except Exception:
            pass


class Stream:
    """Thin RAII wrapper over nts_stream (replaces Cuda_Stream,
    /root/reference/cuda/ntsCUDA.hpp:97-217)."""

    def __init__(self, wrap_hip_stream=None):
        l = lib()
        if wrap_hip_stream is None:
            self.h = l.nts_stream_create()
        else:
            # NOTE: torch's default stream handle is 0 (the HIP null stream);
            # it MUST be wrapped, not replaced — launching on a private
            # non-blocking stream would race torch's fills/copies.
            self.h = l.nts_stream_wrap(_vp(wrap_hip_stream))
        self._lib = l
        _live_streams.add(self)

    @classmethod
    def wrap_torch_current(cls):
        import torch
        return cls(wrap_hip_stream=torch.cuda.current_stream().cuda_stream)

    def sync(self):
        self._lib.nts_stream_sync(self.h)

    def timing(self, enable=True):
        self._lib.nts_stream_timing(self.h, 1 if enable else 0)

    def timing_reset(self):
        self._lib.nts_stream_timing_reset(self.h)

    def kernel_ns(self, tag):
        return self._lib.nts_stream_kernel_ns(self.h, tag)

    def kernel_launches(self, tag):
        return self._lib.nts_stream_kernel_launches(self.h, tag)

    def zero(self, dptr, n_floats):
        self._lib.nts_zero_buffer(self.h, _vp(dptr), n_floats)

    def gather_by_dst_from_src(self, x_ptr, y_ptr, w_ptr, row_indices_ptr,
                               column_offset_ptr, src_s, src_e, dst_s, dst_e,
                               edges, batch, f, with_weight=True):
        self._lib.nts_gather_by_dst_from_src(
            self.h, _vp(x_ptr), _vp(y_ptr), _vp(w_ptr), _vp(row_indices_ptr),
            _vp(column_offset_ptr), src_s, src_e, dst_s, dst_e, edges, batch,
            f, 1 if with_weight else 0)

    def gather_by_src_from_dst(self, g_ptr, y_ptr, w_ptr, row_offset_ptr,
                               column_indices_ptr, src_s, src_e, dst_s, dst_e,
                               edges, batch, f, with_weight=True):
        self._lib.nts_gather_by_src_from_dst(
            self.h, _vp(g_ptr), _vp(y_ptr), _vp(w_ptr), _vp(row_offset_ptr),
            _vp(column_indices_ptr), src_s, src_e, dst_s, dst_e, edges, batch,
            f, 1 if with_weight else 0)

    def deserialize_to_gpu(self, dense_ptr, msg_ptr, count, f, part_s, part_e,
                           sync=False):
        self._lib.nts_deserialize_to_gpu(self.h, _vp(dense_ptr), _vp(msg_ptr),
                                         count, f, part_s, part_e,
                                         1 if sync else 0)

    def aggregate_comm_result(self, master_ptr, msg_ptr, count, f, part_s,
                              part_e, sync=False):
        self._lib.nts_aggregate_comm_result(self.h, _vp(master_ptr),
                                            _vp(msg_ptr), count, f, part_s,
                                            part_e, 1 if sync else 0)

    def gather_rows(self, dense_ptr, packed_ptr, index_ptr, count, row_start, f):
        self._lib.nts_gather_rows(self.h, _vp(dense_ptr), _vp(packed_ptr),
                                  _vp(index_ptr), count, row_start, f)

    def scatter_rows(self, dense_ptr, packed_ptr, index_ptr, count, row_start, f):
        self._lib.nts_scatter_rows(self.h, _vp(dense_ptr), _vp(packed_ptr),
                                   _vp(index_ptr), count, row_start, f)

    def scatter_add_rows(self, dense_ptr, packed_ptr, index_ptr, count,
                         row_start, f):
        self._lib.nts_scatter_add_rows(self.h, _vp(dense_ptr), _vp(packed_ptr),
                                       _vp(index_ptr), count, row_start, f)

    def scatter_src_mirror_to_msg(self, msg, mirror_feat, row_indices,
                                  column_offset, mirror_index, batch, f):
        self._lib.nts_scatter_src_mirror_to_msg(
            self.h, _vp(msg), _vp(mirror_feat), _vp(row_indices),
            _vp(column_offset), _vp(mirror_index), batch, f)

    def gather_msg_to_src_mirror(self, mirror_feat, msg, row_indices,
                                 column_offset, mirror_index, batch, f):
        self._lib.nts_gather_msg_to_src_mirror(
            self.h, _vp(mirror_feat), _vp(msg), _vp(row_indices),
            _vp(column_offset), _vp(mirror_index), batch, f)

    def scatter_dst_to_msg(self, msg, dst_feat, row_indices, column_offset,
                           batch, f):
        self._lib.nts_scatter_dst_to_msg(self.h, _vp(msg), _vp(dst_feat),
                                         _vp(row_indices), _vp(column_offset),
                                         batch, f)

    def gather_msg_to_dst(self, dst_feat, msg, row_indices, column_offset,
                          batch, f):
        self._lib.nts_gather_msg_to_dst(self.h, _vp(dst_feat), _vp(msg),
                                        _vp(row_indices), _vp(column_offset),
                                        batch, f)

    def edge_dot(self, out, dst_rows, src_rows, row_indices, column_offset,
                 src_start, batch, f):
        self._lib.nts_edge_dot(self.h, _vp(out), _vp(dst_rows), _vp(src_rows),
                               _vp(row_indices), _vp(column_offset),
                               src_start, batch, f)

    def scatter_grad_back_to_message(self, input_grad, msg_grad, row_indices,
                                     column_offset, batch, f):
        self._lib.nts_scatter_grad_back_to_message(
            self.h, _vp(input_grad), _vp(msg_grad), _vp(row_indices),
            _vp(column_offset), batch, f)

    def edge_softmax_forward(self, msg_out, msg_in, msg_cached, row_indices,
                             column_offset, batch, f):
        self._lib.nts_edge_softmax_forward(
            self.h, _vp(msg_out), _vp(msg_in), _vp(msg_cached),
            _vp(row_indices), _vp(column_offset), batch, f)

    def edge_softmax_backward(self, msg_in_grad, msg_out_grad, msg_cached,
                              row_indices, column_offset, batch, f):
        self._lib.nts_edge_softmax_backward(
            self.h, _vp(msg_in_grad), _vp(msg_out_grad), _vp(msg_cached),
            _vp(row_indices), _vp(column_offset), batch, f)

    def permute_f32(self, out, inp, index, n):
        self._lib.nts_permute_f32(self.h, _vp(out), _vp(inp), _vp(index), n)

    def edge_softmax_forward_dual(self, out, out_perm, perm_pos, inp, cached,
                                  column_offset, batch, f):
        self._lib.nts_edge_softmax_forward_dual(
            self.h, _vp(out), _vp(out_perm), _vp(perm_pos), _vp(inp),
            _vp(cached), _vp(column_offset), batch, f)

    def edge_softmax_backward_fused(self, in_grad, in_grad_perm, perm_pos,
                                    out_grad, cached, lrelu_input, slope,
                                    column_offset, batch, f, dst_sum=None):
        self._lib.nts_edge_softmax_backward_fused(
            self.h, _vp(in_grad), _vp(in_grad_perm), _vp(perm_pos),
            _vp(out_grad), _vp(cached), _vp(lrelu_input), slope,
            _vp(dst_sum), _vp(column_offset), batch, f)

    def gather_by_src_from_dst_dot(self, inp, out, weight, row_offset,
                                   column_indices, dst_start, batch, edges, f,
                                   dot_vec, dot_out, dot_pos):
        return self._lib.nts_gather_by_src_from_dst_dot(
            self.h, _vp(inp), _vp(out), _vp(weight), _vp(row_offset),
            _vp(column_indices), dst_start, batch, edges, f, _vp(dot_vec),
            _vp(dot_out), _vp(dot_pos))

    def sample_reservoir(self, column_offset, row_indices, dst_list, n_dst,
                         fanout, seed, out_src, out_cnt):
        self._lib.nts_sample_reservoir(self.h, _vp(column_offset),
                                       _vp(row_indices), _vp(dst_list),
                                       n_dst, fanout, seed, _vp(out_src),
                                       _vp(out_cnt))

    def sample_reservoir_dbg_fallback(self, column_offset, row_indices,
                                      dst_list, n_dst, fanout, seed,
                                      out_src, out_cnt):
        """TEST-ONLY: forces the deterministic tie fallback."""
        self._lib.nts_sample_reservoir_dbg_fallback(
            self.h, _vp(column_offset), _vp(row_indices), _vp(dst_list),
            n_dst, fanout, seed, _vp(out_src), _vp(out_cnt))

    def items_cache_clear(self):
        self._lib.nts_items_cache_clear(self.h)

    def items_reuse(self, enable):
        """Opt-in work-item caching; caller must pin its topology buffers."""
        self._lib.nts_items_reuse(self.h, 1 if enable else 0)

    def weight_sum(self, out, weights, offset, batch):
        self._lib.nts_weight_sum(self.h, _vp(out), _vp(weights), _vp(offset),
                                 batch)

    def edge_attention_forward(self, softmax_out, softmax_out_perm, perm_pos,
                               m_sum_out, s_src_mirror, s_dst, row_indices,
                               mirror_index, slope, column_offset, batch):
        self._lib.nts_edge_attention_forward(
            self.h, _vp(softmax_out), _vp(softmax_out_perm), _vp(perm_pos),
            _vp(m_sum_out), _vp(s_src_mirror), _vp(s_dst), _vp(row_indices),
            _vp(mirror_index), slope, _vp(column_offset), batch)

    def destroy(self):
        if self.h:
            self._lib.nts_stream_destroy(self.h)
            self.h = None
