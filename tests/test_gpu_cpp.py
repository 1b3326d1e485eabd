"""The C++ host layer (operator surface over libtorch + C-ABI) runs end to
end on the GPU: parity vs its in-file naive CPU check, then a 2-layer GCN
training loop whose loss must drop (gcn_demo.cpp)."""
import os
import subprocess

import pytest

from tests.conftest import REPO

pytestmark = pytest.mark.gpu

DEMO = os.path.join(REPO, "cpp", "build", "gcn_demo")


def test_gcn_demo_trains():
    if not os.path.exists(DEMO):
        import __graft_entry__
        __graft_entry__._build_cpp()
    env = dict(os.environ)
    env["LD_LIBRARY_PATH"] = (
        os.path.join(REPO, "neutronstarlite_amd") + ":" +
        env.get("LD_LIBRARY_PATH", ""))
    r = subprocess.run([DEMO, "10"], capture_output=True, text=True,
                       timeout=300, env=env)
    assert r.returncode == 0, f"gcn_demo failed:\n{r.stdout}\n{r.stderr}"
    assert "gcn_demo ok" in r.stdout
    assert "PARITY FAIL" not in r.stderr
