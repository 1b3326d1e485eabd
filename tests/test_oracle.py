"""Pins the CPU oracle (oracle/oracle.c) against the committed golden
fixtures (tests/golden/ — values computed by REFERENCE-EXECUTED code via
oracle/_ref, cross-checked by fp64 scipy; see make_golden.py), live
scipy.sparse fp64, the closed form, and record-format roundtrips.  The
reference repo ships no golden vectors of its own for this path
(SURVEY.md §8c); oracle/_ref compiles the reference's sources instead, and
test_ref_parity.py holds oracle and _ref bit-equal."""
import os

import numpy as np
import scipy.sparse as sp

import oracle
from neutronstarlite_amd import graph as G

HERE = os.path.dirname(os.path.abspath(__file__))


def _chunk(edges, w, v):
    return G.build_chunks(edges, w, np.array([0, v], dtype=np.uint32), 0)[0]


def test_cora_golden_all_ones():
    edges = np.load(os.path.join(HERE, "golden", "cora.2708.edge.self.npy"))
    colsum = np.load(os.path.join(HERE, "golden", "cora_w_colsum.f64.npy"))
    v, f = 2708, 4
    outd, ind = oracle.degrees(edges, v)
    w = oracle.norm_weights(np.ascontiguousarray(edges[:, 0]),
                            np.ascontiguousarray(edges[:, 1]), outd, ind)
    ch = _chunk(edges, w, v)
    y = oracle.csc_forward(ch.column_offset, ch.row_indices,
                           ch.edge_weight_forward,
                           np.ones((v, f), np.float32), 0, v, f)
    # closed form: all-ones features -> Y[d] = sum of incident weights
    assert np.allclose(y, colsum[:, None], rtol=1e-5, atol=1e-6)


def test_cora_golden_seeded():
    edges = np.load(os.path.join(HERE, "golden", "cora.2708.edge.self.npy"))
    y_ref = np.load(os.path.join(HERE, "golden", "cora_y_f8.f32.npy"))
    gx_ref = np.load(os.path.join(HERE, "golden", "cora_gx_f8.f32.npy"))
    v, f = 2708, 8
    rng = np.random.default_rng(42)
    x = rng.uniform(-1, 1, size=(v, f)).astype(np.float64)
    g = rng.uniform(-1, 1, size=(v, f)).astype(np.float64)
    outd, ind = oracle.degrees(edges, v)
    w = oracle.norm_weights(np.ascontiguousarray(edges[:, 0]),
                            np.ascontiguousarray(edges[:, 1]), outd, ind)
    ch = _chunk(edges, w, v)
    y = oracle.csc_forward(ch.column_offset, ch.row_indices,
                           ch.edge_weight_forward, x.astype(np.float32), 0, v, f)
    gx = oracle.csr_backward(ch.row_offset, ch.column_indices,
                             ch.edge_weight_backward, g.astype(np.float32), 0, v, f)
    # fixtures are reference-executed (oracle/_ref) fp32 values; the oracle
    # restates the same arithmetic in the same order -> bit-exact
    assert np.array_equal(y, y_ref)
    assert np.array_equal(gx, gx_ref)


def test_forward_backward_vs_scipy_fp64(small_graph):
    v, edges, w = small_graph["v"], small_graph["edges"], small_graph["w"]
    f = 33  # odd width
    rng = np.random.default_rng(1)
    x = rng.uniform(-1, 1, size=(v, f)).astype(np.float32)
    g = rng.uniform(-1, 1, size=(v, f)).astype(np.float32)
    ch = _chunk(edges, w, v)
    A = sp.csr_matrix((w.astype(np.float64), (edges[:, 1], edges[:, 0])),
                      shape=(v, v))
    y = oracle.csc_forward(ch.column_offset, ch.row_indices,
                           ch.edge_weight_forward, x, 0, v, f)
    gx = oracle.csr_backward(ch.row_offset, ch.column_indices,
                             ch.edge_weight_backward, g, 0, v, f)
    assert np.allclose(y, A @ x.astype(np.float64), rtol=2e-4, atol=2e-5)
    assert np.allclose(gx, A.T @ g.astype(np.float64), rtol=2e-4, atol=2e-5)


def test_degree_clamp():
    # isolated vertex 3 gets degree 1, never 0 (graph.hpp:4397-4401)
    edges = np.array([[0, 1], [1, 2], [2, 0]], dtype=np.uint32)
    outd, ind = oracle.degrees(edges, 4)
    assert outd[3] == 1 and ind[3] == 1
    assert outd[0] == 1 and ind[1] == 1


def test_empty_columns_and_self_loop():
    # dst 0 has no in-edges; vertex 2 has a self loop
    edges = np.array([[0, 1], [2, 2], [1, 2]], dtype=np.uint32)
    v, f = 3, 5
    outd, ind = oracle.degrees(edges, v)
    w = oracle.norm_weights(np.ascontiguousarray(edges[:, 0]),
                            np.ascontiguousarray(edges[:, 1]), outd, ind)
    ch = _chunk(edges, w, v)
    x = np.arange(v * f, dtype=np.float32).reshape(v, f)
    y = oracle.csc_forward(ch.column_offset, ch.row_indices,
                           ch.edge_weight_forward, x, 0, v, f)
    assert np.all(y[0] == 0)
    expected_row2 = w[1] * x[2] + w[2] * x[1]
    assert np.allclose(y[2], expected_row2, rtol=1e-6)


def test_message_record_roundtrip():
    """[u32 vid | f x f32] record pack/unpack and partial-sum merge."""
    v, f, n = 64, 6, 20
    rng = np.random.default_rng(3)
    vids = rng.choice(v, size=n, replace=False).astype(np.uint32)
    rows = rng.normal(size=(n, f)).astype(np.float32)
    msg = np.zeros(n * (f + 1), dtype=np.float32)
    rec = msg.reshape(n, f + 1)
    rec[:, 0] = vids.view(np.float32)
    rec[:, 1:] = rows
    dense = np.zeros((v, f), dtype=np.float32)
    oracle.deserialize(msg, n, 0, dense, f)
    assert np.array_equal(dense[vids], rows)
    master = np.ones((v, f), dtype=np.float32)
    oracle.agg_msg_to_master(master, msg, n, 0, f)
    assert np.allclose(master[vids], 1.0 + rows)


def test_oracle_deterministic(small_graph):
    v, edges, w = small_graph["v"], small_graph["edges"], small_graph["w"]
    ch = _chunk(edges, w, v)
    x = np.random.default_rng(5).normal(size=(v, 17)).astype(np.float32)
    y1 = oracle.csc_forward(ch.column_offset, ch.row_indices,
                            ch.edge_weight_forward, x, 0, v, 17)
    y2 = oracle.csc_forward(ch.column_offset, ch.row_indices,
                            ch.edge_weight_forward, x, 0, v, 17)
    assert np.array_equal(y1, y2)
