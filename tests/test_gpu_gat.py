"""GPU parity for the GAT edge-valued kernels (BASELINE config #5) vs the
CPU oracle: scatter/gather between vertex rows and per-edge messages, and
the per-destination edge softmax."""
import numpy as np
import pytest
import torch

import oracle
from neutronstarlite_amd import graph as G

pytestmark = pytest.mark.gpu

RTOL, ATOL = 1e-4, 1e-5


def assert_close(got, ref, name=""):
    got = got.cpu().numpy()
    err = np.abs(got - ref)
    bad = err > RTOL * np.abs(ref) + ATOL
    assert not bad.any(), f"{name}: {bad.sum()}/{bad.size} out of tol"


@pytest.fixture(scope="module")
def setup():
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    v, e = 1500, 30000
    edges = G.rmat_edges(v, e, seed=11)
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    ch = G.build_chunks(edges, w, np.array([0, v], dtype=np.uint32), 0)[0]
    uniq = np.unique(ch.row_indices)
    mi = np.zeros(v, dtype=np.uint32)
    mi[uniq] = np.arange(len(uniq), dtype=np.uint32)
    from neutronstarlite_amd import shim
    s = shim.Stream.wrap_torch_current()

    def up32(a):
        return torch.from_numpy(np.ascontiguousarray(a).view(np.int32)).to(dev)

    return {"dev": dev, "v": v, "ch": ch, "uniq": uniq, "mi": mi, "s": s,
            "d_coff": up32(ch.column_offset), "d_rows": up32(ch.row_indices),
            "d_mi": up32(mi)}


def test_scatter_src_and_gather_back(setup):
    st, ch, dev = setup["s"], setup["ch"], setup["dev"]
    f, E, M = 16, ch.edge_size, len(setup["uniq"])
    rng = np.random.default_rng(0)
    mirror = rng.normal(size=(M, f)).astype(np.float32)
    msg_ref = np.zeros((E, f), dtype=np.float32)
    oracle.scatter_src_to_msg(msg_ref, mirror, ch.row_indices,
                              ch.column_offset, setup["mi"], ch.dst_n, f)
    mt = torch.from_numpy(mirror).to(dev)
    msg = torch.zeros(E, f, device=dev)
    st.scatter_src_mirror_to_msg(msg.data_ptr(), mt.data_ptr(),
                                 setup["d_rows"].data_ptr(),
                                 setup["d_coff"].data_ptr(),
                                 setup["d_mi"].data_ptr(), ch.dst_n, f)
    torch.cuda.synchronize()
    assert_close(msg, msg_ref, "scatter_src")

    back_ref = np.zeros((M, f), dtype=np.float32)
    oracle.gather_msg_to_src(back_ref, msg_ref, ch.row_indices,
                             ch.column_offset, setup["mi"], ch.dst_n, f)
    back = torch.zeros(M, f, device=dev)
    st.gather_msg_to_src_mirror(back.data_ptr(), msg.data_ptr(),
                                setup["d_rows"].data_ptr(),
                                setup["d_coff"].data_ptr(),
                                setup["d_mi"].data_ptr(), ch.dst_n, f)
    torch.cuda.synchronize()
    assert_close(back, back_ref, "gather_src")


def test_scatter_dst_gather_dst_and_grad(setup):
    st, ch, dev = setup["s"], setup["ch"], setup["dev"]
    f, E = 8, ch.edge_size
    rng = np.random.default_rng(1)
    dstf = rng.normal(size=(ch.dst_n, f)).astype(np.float32)
    msg_ref = np.zeros((E, f), dtype=np.float32)
    oracle.scatter_dst_to_msg(msg_ref, dstf, ch.column_offset, ch.dst_n, f)
    dt = torch.from_numpy(dstf).to(dev)
    msg = torch.zeros(E, f, device=dev)
    st.scatter_dst_to_msg(msg.data_ptr(), dt.data_ptr(),
                          setup["d_rows"].data_ptr(),
                          setup["d_coff"].data_ptr(), ch.dst_n, f)
    torch.cuda.synchronize()
    assert_close(msg, msg_ref, "scatter_dst")

    acc_ref = np.zeros_like(dstf)
    oracle.gather_msg_to_dst(acc_ref, msg_ref, ch.column_offset, ch.dst_n, f)
    acc = torch.zeros(ch.dst_n, f, device=dev)
    st.gather_msg_to_dst(acc.data_ptr(), msg.data_ptr(),
                         setup["d_rows"].data_ptr(),
                         setup["d_coff"].data_ptr(), ch.dst_n, f)
    torch.cuda.synchronize()
    assert_close(acc, acc_ref, "gather_dst")

    mg_ref = np.zeros((E, f), dtype=np.float32)
    oracle.scatter_grad_back_to_msg(dstf, mg_ref, ch.column_offset, ch.dst_n, f)
    mg = torch.zeros(E, f, device=dev)
    st.scatter_grad_back_to_message(dt.data_ptr(), mg.data_ptr(),
                                    setup["d_rows"].data_ptr(),
                                    setup["d_coff"].data_ptr(), ch.dst_n, f)
    torch.cuda.synchronize()
    assert_close(mg, mg_ref, "scatter_grad_back")


@pytest.mark.parametrize("f", [1, 4])
def test_edge_softmax_fwd_bwd(setup, f):
    st, ch, dev = setup["s"], setup["ch"], setup["dev"]
    E = ch.edge_size
    rng = np.random.default_rng(2)
    scores = rng.normal(scale=2.0, size=(E, f)).astype(np.float32)
    out_ref = np.zeros_like(scores)
    cached_ref = np.zeros_like(scores)
    oracle.edge_softmax_forward(out_ref, scores, cached_ref, ch.column_offset,
                                ch.dst_n, f)
    it = torch.from_numpy(scores).to(dev)
    out = torch.zeros(E, f, device=dev)
    cached = torch.zeros(E, f, device=dev)
    st.edge_softmax_forward(out.data_ptr(), it.data_ptr(), cached.data_ptr(),
                            setup["d_rows"].data_ptr(),
                            setup["d_coff"].data_ptr(), ch.dst_n, f)
    torch.cuda.synchronize()
    assert_close(out, out_ref, "softmax fwd")
    assert_close(cached, cached_ref, "softmax cached")

    g = rng.normal(size=(E, f)).astype(np.float32)
    gin_ref = np.zeros_like(scores)
    oracle.edge_softmax_backward(gin_ref, g, cached_ref, ch.column_offset,
                                 ch.dst_n, f)
    gt = torch.from_numpy(g).to(dev)
    gin = torch.zeros(E, f, device=dev)
    st.edge_softmax_backward(gin.data_ptr(), gt.data_ptr(), cached.data_ptr(),
                             setup["d_rows"].data_ptr(),
                             setup["d_coff"].data_ptr(), ch.dst_n, f)
    torch.cuda.synchronize()
    assert_close(gin, gin_ref, "softmax bwd")


def test_fused_attention_forward_matches_decomposed(setup):
    """nts_edge_attention_forward (one pass for scatter_src + scatter_dst +
    leaky_relu + exp + sums, dual-order softmax emission) vs the decomposed
    kernel chain it replaces, and nts_edge_softmax_forward_dual vs the plain
    softmax + permute."""
    st, ch, dev = setup["s"], setup["ch"], setup["dev"]
    E, M = ch.edge_size, len(setup["uniq"])
    slope = 0.2
    rng = np.random.default_rng(9)
    s_src = torch.from_numpy(rng.normal(size=(M, 1)).astype(np.float32)).to(dev)
    s_dst = torch.from_numpy(
        rng.normal(size=(ch.dst_n, 1)).astype(np.float32)).to(dev)
    # decomposed chain
    m_src = torch.empty(E, 1, device=dev)
    m_dst = torch.empty(E, 1, device=dev)
    st.scatter_src_mirror_to_msg(m_src.data_ptr(), s_src.data_ptr(),
                                 setup["d_rows"].data_ptr(),
                                 setup["d_coff"].data_ptr(),
                                 setup["d_mi"].data_ptr(), ch.dst_n, 1)
    st.scatter_dst_to_msg(m_dst.data_ptr(), s_dst.data_ptr(),
                          setup["d_rows"].data_ptr(),
                          setup["d_coff"].data_ptr(), ch.dst_n, 1)
    m_sum_ref = m_src + m_dst
    e_val = torch.nn.functional.leaky_relu(m_sum_ref, slope).contiguous()
    s_ref = torch.empty(E, 1, device=dev)
    cached_ref = torch.empty(E, 1, device=dev)
    st.edge_softmax_forward(s_ref.data_ptr(), e_val.data_ptr(),
                            cached_ref.data_ptr(), setup["d_rows"].data_ptr(),
                            setup["d_coff"].data_ptr(), ch.dst_n, 1)
    # permutation map (CSC -> CSR)
    perm = np.argsort(ch.row_indices.astype(np.int64), kind="stable")
    inv = np.empty_like(perm)
    inv[perm] = np.arange(len(perm))
    d_inv = torch.from_numpy(inv.astype(np.uint32).view(np.int32)).to(dev)
    # fused
    s_f = torch.empty(E, 1, device=dev)
    s_f_csr = torch.empty(E, 1, device=dev)
    m_sum_f = torch.empty(E, 1, device=dev)
    st.edge_attention_forward(s_f.data_ptr(), s_f_csr.data_ptr(),
                              d_inv.data_ptr(), m_sum_f.data_ptr(),
                              s_src.data_ptr(), s_dst.data_ptr(),
                              setup["d_rows"].data_ptr(),
                              setup["d_mi"].data_ptr(), slope,
                              setup["d_coff"].data_ptr(), ch.dst_n)
    torch.cuda.synchronize()
    assert_close(m_sum_f, m_sum_ref.cpu().numpy(), "fused m_sum")
    assert_close(s_f, s_ref.cpu().numpy(), "fused softmax")
    assert_close(s_f_csr, s_ref.cpu().numpy()[perm], "fused softmax CSR order")
    # the dual-emitting softmax entry alone
    s_d = torch.empty(E, 1, device=dev)
    s_d_csr = torch.empty(E, 1, device=dev)
    cc = torch.empty(E, 1, device=dev)
    st.edge_softmax_forward_dual(s_d.data_ptr(), s_d_csr.data_ptr(),
                                 d_inv.data_ptr(), e_val.data_ptr(),
                                 cc.data_ptr(), setup["d_coff"].data_ptr(),
                                 ch.dst_n, 1)
    torch.cuda.synchronize()
    assert_close(s_d, s_ref.cpu().numpy(), "dual softmax")
    assert_close(s_d_csr, s_ref.cpu().numpy()[perm], "dual softmax CSR")
