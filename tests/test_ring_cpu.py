"""Multi-process ring-exchange logic under gloo (world_size 2, CPU): the
distributed path of ring.py must reproduce the whole-graph oracle result.
The aggregation engine is oracle-backed HERE ONLY (test infrastructure);
the product engine is HIP (ops.HipEngine) and is covered by the gpu tests."""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

import oracle
from neutronstarlite_amd import graph as G
from neutronstarlite_amd.ring import RingGraph, ring_backward, ring_forward

V, E, F, SEED = 777, 9000, 11, 7


class OracleEngine:
    """Test-only CPU engine with the engine protocol ring.py expects."""

    def csc_forward(self, ch, x_block, y, with_weight=True):
        assert with_weight
        oracle.csc_forward(ch.column_offset, ch.row_indices,
                           ch.edge_weight_forward, x_block.numpy(), ch.src_s,
                           ch.dst_n, y.shape[1], out=y.numpy())

    def csr_backward(self, ch, grad_block, out, with_weight=True):
        assert with_weight
        oracle.csr_backward(ch.row_offset, ch.column_indices,
                            ch.edge_weight_backward, grad_block.numpy(),
                            ch.dst_s, ch.src_n, out.shape[1], out=out.numpy())


def _data(world):
    edges = G.rmat_edges(V, E, seed=SEED)
    outd, ind = G.degrees(edges, V)
    w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
    offs = G.partition_offsets(edges, V, world)
    rng = np.random.default_rng(42)
    x = rng.uniform(-1, 1, size=(V, F)).astype(np.float32)
    g = rng.uniform(-1, 1, size=(V, F)).astype(np.float32)
    return edges, w, offs, x, g


def _worker(rank, world, filtered, tmpdir, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        dist.init_process_group(
            "gloo", init_method=f"file://{tmpdir}/pg", rank=rank,
            world_size=world)
        edges, w, offs, x, g = _data(world)
        chunks = G.build_chunks(edges, w, offs, rank)
        rg = RingGraph(offs, rank, chunks, torch.device("cpu"))
        if filtered:
            from neutronstarlite_amd.ring import setup_mirror_lists
            setup_mirror_lists(rg)
        lo, hi = int(offs[rank]), int(offs[rank + 1])
        eng = OracleEngine()
        y = ring_forward(rg, torch.from_numpy(x[lo:hi]).clone(), eng)
        gx = ring_backward(rg, torch.from_numpy(g[lo:hi]).clone(), eng)
        q.put((rank, y.numpy(), gx.numpy()))
        dist.destroy_process_group()
    except Exception as exc:  # surface worker failures to the main process
        q.put((rank, "error", repr(exc)))
        raise


@pytest.mark.timeout(300)
@pytest.mark.parametrize("world,filtered", [(2, False), (3, False), (3, True)])
def test_ring_matches_whole_graph(tmp_path, world, filtered):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker,
                         args=(r, world, filtered, str(tmp_path), q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, y, gx = q.get()
        assert not (isinstance(y, str) and y == "error"), f"rank {rank}: {gx}"
        results[rank] = (y, gx)
    for p in procs:
        p.join(60)
        assert p.exitcode == 0

    edges, w, offs, x, g = _data(world)
    ch = G.build_chunks(edges, w, np.array([0, V], dtype=np.uint32), 0)[0]
    y_ref = oracle.csc_forward(ch.column_offset, ch.row_indices,
                               ch.edge_weight_forward, x, 0, V, F)
    gx_ref = oracle.csr_backward(ch.row_offset, ch.column_indices,
                                 ch.edge_weight_backward, g, 0, V, F)
    y_all = np.concatenate([results[r][0] for r in range(world)])
    gx_all = np.concatenate([results[r][1] for r in range(world)])
    assert np.allclose(y_all, y_ref, rtol=1e-4, atol=1e-5)
    assert np.allclose(gx_all, gx_ref, rtol=1e-4, atol=1e-5)


def _worker_no_cross(rank, world, tmpdir, q):
    """Partitions with NO cross-partition edges: the mirror-filtered ring's
    need/serve lists are all empty, exercising the skip-empty-P2P logic
     end to end under gloo."""
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        dist.init_process_group("gloo", init_method=f"file://{tmpdir}/pg2",
                                rank=rank, world_size=world)
        v_half = 100
        v = v_half * world
        rng = np.random.default_rng(4)
        # edges strictly inside each contiguous block of 100 vertices
        src, dst = [], []
        for b in range(world):
            s = rng.integers(b * v_half, (b + 1) * v_half, 600)
            d = rng.integers(b * v_half, (b + 1) * v_half, 600)
            src.append(s)
            dst.append(d)
        edges = np.stack([np.concatenate(src), np.concatenate(dst)],
                         axis=1).astype(np.uint32)
        outd, ind = G.degrees(edges, v)
        w = G.norm_weights(edges[:, 0], edges[:, 1], outd, ind)
        offs = np.arange(world + 1, dtype=np.uint32) * v_half
        chunks = G.build_chunks(edges, w, offs, rank)
        rg = RingGraph(offs, rank, chunks, torch.device("cpu"))
        from neutronstarlite_amd.ring import setup_mirror_lists
        setup_mirror_lists(rg)
        for k in range(world):
            if k != rank:
                assert len(rg.need[k]) == 0 and len(rg.serve[k]) == 0
        f = 5
        x = rng.uniform(-1, 1, size=(v, f)).astype(np.float32)
        lo, hi = int(offs[rank]), int(offs[rank + 1])
        eng = OracleEngine()
        y = ring_forward(rg, torch.from_numpy(x[lo:hi]).clone(), eng)
        q.put((rank, y.numpy(), x))
        dist.destroy_process_group()
    except Exception as exc:
        q.put((rank, "error", repr(exc)))
        raise


@pytest.mark.timeout(300)
def test_ring_no_cross_partition_edges(tmp_path):
    world = 3
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker_no_cross,
                         args=(r, world, str(tmp_path), q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, y, x = q.get()
        assert not (isinstance(y, str) and y == "error"), f"rank {rank}: {x}"
        results[rank] = (y, x)
    for p in procs:
        p.join(60)
        assert p.exitcode == 0
    # whole-graph oracle check
    v_half, v, f = 100, 300, 5
    _, x = results[0]
    edges_chk = None  # per-rank graphs are identical (seeded)
    # reconstruct from rank 0's seeded generation
    rng = np.random.default_rng(4)
    src, dst = [], []
    for b in range(world):
        src.append(rng.integers(b * v_half, (b + 1) * v_half, 600))
        dst.append(rng.integers(b * v_half, (b + 1) * v_half, 600))
    edges_chk = np.stack([np.concatenate(src), np.concatenate(dst)],
                         axis=1).astype(np.uint32)
    outd, ind = G.degrees(edges_chk, v)
    w = G.norm_weights(edges_chk[:, 0], edges_chk[:, 1], outd, ind)
    ch = G.build_chunks(edges_chk, w, np.array([0, v], dtype=np.uint32), 0)[0]
    y_ref = oracle.csc_forward(ch.column_offset, ch.row_indices,
                               ch.edge_weight_forward, x, 0, v, f)
    y_all = np.concatenate([results[r][0] for r in range(world)])
    assert np.allclose(y_all, y_ref, rtol=1e-4, atol=1e-5)
