"""Generates the committed golden fixtures under tests/golden/.

Run in the build container (where /root/reference is mounted); the GPU box
only reads the committed .npy files.  Fixtures:

  cora.2708.edge.self.npy   the vendored Cora edge list
                            (/root/reference/data/cora.2708.edge.self,
                            Gemini binary: (src,dst) u32 pairs, 8 B/edge —
                            a public dataset file, not reference code)
  cora_w_colsum.f64.npy     per-destination sum of norm-degree weights
                            (closed form: all-ones features give
                            Y[d] = sum_e w_e), computed in float64 with an
                            implementation independent of oracle/ (numpy
                            bincount over fp64 weights)
  cora_y_f8.f32.npy         forward aggregation output, f=8 seeded U(-1,1)
                            features (seed 42), computed by REFERENCE-EXECUTED
                            code: ForwardCPUfuseOp::forward compiled from
                            /root/reference/core/ntsCPUFusedGraphOp.hpp:41-109
                            by oracle/ref_harness (oracle/_ref, parity kind
                            "reference"), cross-checked here against fp64
                            scipy.sparse within 2e-4 rel
  cora_gx_f8.f32.npy        backward (CSR) output, same setup, via
                            ForwardCPUfuseOp::backward
                            (ntsCPUFusedGraphOp.hpp:110-167)

The committed values are therefore anchored to code the reference itself
authored; tests/test_oracle.py additionally requires oracle/oracle.c to
match them BIT-EXACTLY, and tests/test_ref_parity.py holds the oracle and
_ref bit-equal on random graphs.
"""
import os
import sys

import numpy as np
import scipy.sparse as sp

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(os.path.dirname(HERE))
sys.path.insert(0, REPO)

from neutronstarlite_amd import graph as G  # noqa: E402
import oracle.ref as ref                     # noqa: E402

REF_EDGE = "/root/reference/data/cora.2708.edge.self"
V, F = 2708, 8


def main():
    edges = G.load_gemini_edges(REF_EDGE)
    assert edges.shape == (13566, 2) and edges.max() < V
    np.save(os.path.join(HERE, "cora.2708.edge.self.npy"), edges)

    outd = np.maximum(np.bincount(edges[:, 0], minlength=V), 1)
    ind = np.maximum(np.bincount(edges[:, 1], minlength=V), 1)
    w64 = 1.0 / (np.sqrt(outd[edges[:, 0]].astype(np.float64)) *
                 np.sqrt(ind[edges[:, 1]].astype(np.float64)))
    colsum = np.zeros(V)
    np.add.at(colsum, edges[:, 1], w64)
    np.save(os.path.join(HERE, "cora_w_colsum.f64.npy"), colsum)

    rng = np.random.default_rng(42)
    x = rng.uniform(-1, 1, size=(V, F))
    g = rng.uniform(-1, 1, size=(V, F))

    # reference-executed values (oracle/_ref): the committed fixtures
    assert ref.available(), "oracle/_ref must build here (reference mounted)"
    outd32, ind32 = G.degrees(edges, V)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd32, ind32)
    ch = G.build_chunks(edges, w, np.array([0, V], dtype=np.uint32), 0)[0]
    y = ref.fused_forward(V, F, ch.column_offset, ch.row_indices,
                          ch.row_offset, ch.column_indices,
                          outd32.astype(np.uint32), ind32.astype(np.uint32),
                          x.astype(np.float32))
    gx = ref.fused_backward(V, F, ch.column_offset, ch.row_indices,
                            ch.row_offset, ch.column_indices,
                            outd32.astype(np.uint32), ind32.astype(np.uint32),
                            g.astype(np.float32))

    # independent fp64 scipy cross-check of the reference-executed values
    A = sp.csr_matrix((w64, (edges[:, 1], edges[:, 0])), shape=(V, V))
    assert np.allclose(y, A @ x, rtol=2e-4, atol=2e-5)
    assert np.allclose(gx, A.T @ g, rtol=2e-4, atol=2e-5)

    np.save(os.path.join(HERE, "cora_y_f8.f32.npy"), y)
    np.save(os.path.join(HERE, "cora_gx_f8.f32.npy"), gx)
    print("golden fixtures written (values: reference-executed oracle/_ref)")


if __name__ == "__main__":
    main()
