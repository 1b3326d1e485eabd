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
                            features (seed 42), computed with fp64
                            scipy.sparse — independent of oracle/ and of the
                            HIP kernels — rounded to fp32
  cora_gx_f8.f32.npy        backward (CSR) output, same setup, A^T pull
"""
import os
import sys

import numpy as np
import scipy.sparse as sp

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(os.path.dirname(HERE))
sys.path.insert(0, REPO)

from neutronstarlite_amd import graph as G  # noqa: E402

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
    A = sp.csr_matrix((w64, (edges[:, 1], edges[:, 0])), shape=(V, V))
    np.save(os.path.join(HERE, "cora_y_f8.f32.npy"), (A @ x).astype(np.float32))
    np.save(os.path.join(HERE, "cora_gx_f8.f32.npy"), (A.T @ g).astype(np.float32))
    print("golden fixtures written")


if __name__ == "__main__":
    main()
