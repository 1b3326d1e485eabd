"""neutronstarlite_amd — MI355X-native rebuild of NeutronStarLite's GNN
neighbor-aggregation hot path (see SURVEY.md §8 and DESIGN.md).

Layout:
  csrc/nts_hip.hip  hand-written gfx950 HIP kernels + the C-ABI shim
                    (include/nts_hip.h) — the product compute path
  shim.py           ctypes binding of that C-ABI (fails loudly if unbuilt)
  graph.py          host-side graph loading / partitioning / chunk building
  ops.py            the operator layer mirroring the reference's
                    ForwardSingleGPUfuseOp / ForwardGPUfuseOp surface
  ring.py           RCCL (torch.distributed) ring mirror exchange
"""
__version__ = "0.1.0"
