"""Guards the committed golden fixtures against drift: re-running the
generating script (which reads /root/reference) must reproduce the committed
files bit-exactly.  Skipped where the reference is not mounted (GPU boxes)."""
import os

import numpy as np
import pytest

from tests.conftest import REFERENCE, REPO

GOLDEN = os.path.join(REPO, "tests", "golden")


@pytest.mark.skipif(not os.path.exists(REFERENCE),
                    reason="reference not mounted (run-time box)")
def test_make_golden_reproduces_committed(tmp_path, monkeypatch):
    import tests.golden.make_golden as mg
    work = tmp_path / "golden"
    work.mkdir()
    monkeypatch.setattr(mg, "HERE", str(work))
    mg.main()
    for fn in ("cora.2708.edge.self.npy", "cora_w_colsum.f64.npy",
               "cora_y_f8.f32.npy", "cora_gx_f8.f32.npy"):
        a = np.load(os.path.join(GOLDEN, fn))
        b = np.load(str(work / fn))
        assert np.array_equal(a, b), f"fixture drift: {fn}"
