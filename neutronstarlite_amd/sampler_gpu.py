"""GPU-resident mini-batch sampling (SURVEY §8f-3 "GPU-resident sampling"):
the reservoir fan-out selection runs as a gfx950 kernel over the device CSC
(`nts_sample_reservoir`, reservoir semantics of ntsSampler.hpp:113-166);
the sampCSC-style compaction (coocsc.hpp:62-89) is torch index plumbing on
device — nothing touches the host until the caller asks.

Differences vs the host sampler (sampler.py), both within the reference's
contract (<= fanout uniformly chosen in-neighbor slots per destination):
  - the RNG sequence is a counter hash (deterministic in seed/dst/step);
  - compacted local source ids are in SORTED-global order (torch.unique)
    rather than first-occurrence order — the mapping is carried explicitly
    in `src`, so downstream arithmetic is order-independent.
"""
from dataclasses import dataclass

import torch

from . import shim


@dataclass
class GpuSampledLayer:
    """Device-resident compacted subgraph (tensors on the sampling GPU)."""
    dst: torch.Tensor               # int32 [n_dst] global dst ids
    src: torch.Tensor               # int64 [n_src] global src ids (sorted)
    column_offset: torch.Tensor     # int32 [n_dst+1]
    row_indices_local: torch.Tensor  # int32 [E']
    edge_weight: torch.Tensor       # f32 [E']
    row_offset: torch.Tensor        # int32 [n_src+1]
    column_indices_local: torch.Tensor  # int32 [E']
    edge_weight_backward: torch.Tensor  # f32 [E']

    @property
    def n_dst(self):
        return int(self.dst.numel())

    @property
    def n_src(self):
        return int(self.src.numel())

    @property
    def e_size(self):
        return int(self.row_indices_local.numel())


def sample_layer_gpu(stream: shim.Stream, d_column_offset: torch.Tensor,
                     d_row_indices: torch.Tensor, dst_list: torch.Tensor,
                     fanout: int, d_outd: torch.Tensor, d_ind: torch.Tensor,
                     seed: int) -> GpuSampledLayer:
    """One layer of fan-out sampling fully on device.

    d_column_offset/d_row_indices: whole-graph CSC (int32-viewed u32, cuda);
    dst_list: int32 cuda global destination ids; d_outd/d_ind: full-graph
    degrees (int32 cuda, clamped >=1) for the norm-degree weights."""
    dev = dst_list.device
    n = int(dst_list.numel())
    out_src = torch.zeros(n * fanout, dtype=torch.int32, device=dev)
    out_cnt = torch.zeros(n, dtype=torch.int32, device=dev)
    stream.sample_reservoir(d_column_offset.data_ptr(),
                            d_row_indices.data_ptr(), dst_list.data_ptr(),
                            n, fanout, seed, out_src.data_ptr(),
                            out_cnt.data_ptr())
    cnt = out_cnt.to(torch.int64)
    col_off = torch.zeros(n + 1, dtype=torch.int64, device=dev)
    col_off[1:] = torch.cumsum(cnt, 0)
    # compaction with exactly TWO host syncs (E and n_src): boolean-mask
    # selection and torch.unique each forced an extra device->host sync per
    # layer, which dominated the sampled step once the kernels got fast
    e_sz = int(col_off[-1].item())                       # sync 1
    d_local = torch.repeat_interleave(
        torch.arange(n, device=dev), cnt, output_size=e_sz)
    j_within = torch.arange(e_sz, device=dev) - col_off[d_local]
    # dst-major order == the local CSC order
    src_g = (out_src.view(-1).to(torch.int64)[d_local * fanout + j_within]
             & 0xFFFFFFFF)
    # sorted-unique + inverse via sort/flags (no torch.unique sync)
    sorted_src, sort_idx = torch.sort(src_g)
    new_flag = torch.ones(e_sz, dtype=torch.bool, device=dev)
    if e_sz > 1:
        new_flag[1:] = sorted_src[1:] != sorted_src[:-1]
    ril_sorted = torch.cumsum(new_flag, 0) - 1
    ril = torch.empty_like(ril_sorted)
    ril[sort_idx] = ril_sorted
    n_uniq = int(ril_sorted[-1].item()) + 1 if e_sz else 0   # sync 2
    src_unique = torch.zeros(n_uniq, dtype=torch.int64, device=dev)
    src_unique[ril_sorted] = sorted_src   # duplicate writes carry equal values
    dst_g = (dst_list.to(torch.int64) & 0xFFFFFFFF)[d_local]
    w = (1.0 / (torch.sqrt(d_outd[src_g].float()) *
                torch.sqrt(d_ind[dst_g].float())))
    # local CSR = stable-by-src permutation of the CSC order
    perm = torch.argsort(ril, stable=True)
    n_src = int(src_unique.numel())
    row_off = torch.zeros(n_src + 1, dtype=torch.int64, device=dev)
    row_off[1:] = torch.cumsum(torch.bincount(ril[perm], minlength=n_src), 0)
    return GpuSampledLayer(
        dst=dst_list,
        src=src_unique,
        column_offset=col_off.to(torch.int32),
        row_indices_local=ril.to(torch.int32),
        edge_weight=w.contiguous(),
        row_offset=row_off.to(torch.int32),
        column_indices_local=d_local[perm].to(torch.int32),
        edge_weight_backward=w[perm].contiguous(),
    )


def sample_subgraph_gpu(stream, d_column_offset, d_row_indices, targets,
                        fanouts, d_outd, d_ind, seed=0):
    """Layer-wise chain on device: layer i's destinations are layer i-1's
    compacted sources (ntsSampler.hpp layer loop)."""
    layers = []
    dst = targets
    for i, f in enumerate(fanouts):
        ly = sample_layer_gpu(stream, d_column_offset, d_row_indices, dst,
                              int(f), d_outd, d_ind, seed + i)
        layers.append(ly)
        dst = ly.src.to(torch.int32)
    return layers
