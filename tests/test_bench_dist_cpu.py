"""Drives bench.py's EXACT multi-rank orchestration (env parsing, process
group, ring steps, barriers, max-over-ranks timing, JSON contract) under
gloo on CPU with a test-owned oracle engine — the code path the driver's
round-end SCALE run takes on 8 GPUs, minus RCCL and the HIP kernels."""
import io
import json
import os
import sys

import numpy as np
import pytest
import torch.multiprocessing as mp

from tests.conftest import REPO


class _NullStream:
    def timing(self, *a):
        pass

    def timing_reset(self):
        pass

    def kernel_ns(self, tag):
        return 0.0

    def kernel_launches(self, tag):
        return 0


class _OracleEngine:
    """Engine protocol over numpy Chunks on CPU tensors (test only)."""

    def __init__(self):
        self.stream = _NullStream()

    def csc_forward(self, ch, x_block, y, with_weight=True):
        import oracle
        oracle.csc_forward(ch.column_offset, ch.row_indices,
                           ch.edge_weight_forward, x_block.numpy(), ch.src_s,
                           ch.dst_n, y.shape[1], out=y.numpy())

    def csr_backward(self, ch, grad_block, out, with_weight=True):
        import oracle
        oracle.csr_backward(ch.row_offset, ch.column_indices,
                            ch.edge_weight_backward, grad_block.numpy(),
                            ch.dst_s, ch.src_n, out.shape[1], out=out.numpy())


def _worker(rank, world, port, flags, q):
    try:
        os.environ.update({
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
            "WORLD_SIZE": str(world), "RANK": str(rank),
            "LOCAL_RANK": str(rank),
        })
        sys.path.insert(0, REPO)
        import bench
        buf = io.StringIO()
        stdout = sys.stdout
        sys.stdout = buf
        try:
            bench.main(
                ["--graph", "small", "--feat", "24", "--steps", "2",
                 "--warmup", "1", "--no-cpu-baseline", "--gpus", str(world)]
                + flags,
                _test_engine_factory=_OracleEngine,
                _test_backend="gloo", _test_device="cpu")
        finally:
            sys.stdout = stdout
        q.put((rank, buf.getvalue()))
    except Exception as exc:
        q.put((rank, f"ERROR: {exc!r}"))
        raise


@pytest.mark.timeout(420)
@pytest.mark.parametrize("world,flags", [(3, []), (3, ["--mirror-filtered"]),
                                         (4, []), (4, ["--mirror-filtered"])])
def test_bench_orchestration(world, flags):
    # world 4 runs the 8-GPU SCALE shape's ring structure (3 exchange steps
    # per direction, pipelined backward double-buffering) under gloo
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29611 + len(flags) + world * 7
    procs = [ctx.Process(target=_worker, args=(r, world, port, flags, q))
             for r in range(world)]
    for p in procs:
        p.start()
    outs = {}
    for _ in range(world):
        r, out = q.get()
        assert not str(out).startswith("ERROR"), f"rank {r}: {out}"
        outs[r] = out
    for p in procs:
        p.join(60)
        assert p.exitcode == 0
    # exactly rank 0 prints exactly one JSON line with the contract fields
    lines = [l for l in outs[0].splitlines() if l.startswith("{")]
    assert len(lines) == 1
    for r in range(1, world):
        assert not [l for l in outs[r].splitlines() if l.startswith("{")]
    d = json.loads(lines[0])
    assert d["metric"] == "aggregated_edges_per_sec"
    assert d["n_gpus"] == world and d["steps"] == 2 and d["warmup"] == 1
    assert d["scaling"] == "strong" and d["dtype"] == "f32"
    assert d["config"]["parallelism"] == f"graph-partitioned dp{world}"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert "roofline" in d and "cpu_baseline" in d
