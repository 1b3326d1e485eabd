"""End-to-end: bench.py consumes a reference-format cfg + Gemini edge file
(Cora, written from the committed fixture) and produces a valid JSON line."""
import json
import os
import subprocess
import sys

import numpy as np
import pytest

from tests.conftest import REPO

pytestmark = pytest.mark.gpu


@pytest.mark.timeout(300)
def test_bench_runs_reference_cfg(tmp_path):
    edges = np.load(os.path.join(REPO, "tests", "golden",
                                 "cora.2708.edge.self.npy"))
    edge_path = tmp_path / "cora.2708.edge.self"
    edges.astype(np.uint32).tofile(edge_path)
    cfg = tmp_path / "gcn_cora.cfg"
    cfg.write_text(
        "ALGORITHM:GCN\nVERTICES:2708\nLAYERS:1433-128-7\nEPOCHS:10\n"
        f"EDGE_FILE:{edge_path}\nPROC_CUDA:1\n")
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--cfg", str(cfg),
         "--steps", "3", "--warmup", "1", "--no-cpu-baseline"],
        capture_output=True, text=True, timeout=280, cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["metric"] == "aggregated_edges_per_sec"
    assert d["value"] > 0
    assert d["config"]["V"] == 2708 and d["config"]["feat"] == 1433
    assert d["config"]["E"] == 13566
