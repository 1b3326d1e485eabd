"""North_star link contract for the flagship dist-GCN toolkit: the per-layer
loop of toolkits/GCN.hpp:217-235 (runGraphOp<ForwardGPUfuseOp> + two-input
runVertexForward), vendored into cpp/gcn_link_check.cpp, compiles and links
against cpp/include/nts/nts.hpp.  On CPU the binary exits 0 after the link
proof; on a GPU (test_gpu_rccl.py) it additionally runs the loop at world-1
RCCL."""
import os
import subprocess

from tests.conftest import REPO

BIN = os.path.join(REPO, "cpp", "build", "gcn_link_check")


def _env():
    env = dict(os.environ)
    env["LD_LIBRARY_PATH"] = (
        os.path.join(REPO, "neutronstarlite_amd") + ":" +
        os.path.join(REPO, "cpp", "build") + ":" +
        env.get("LD_LIBRARY_PATH", ""))
    return env


def test_flagship_loop_compiles_and_links():
    if not os.path.exists(BIN):
        import __graft_entry__
        __graft_entry__._build_cpp()
    assert os.path.exists(BIN), "gcn_link_check did not build"
    r = subprocess.run([BIN], capture_output=True, text=True, timeout=300,
                       env=_env())
    assert r.returncode == 0, f"gcn_link_check failed:\n{r.stdout}\n{r.stderr}"


def test_host_layer_units_cpu():
    """CPU-runnable unit checks of the C++ host layer: Parameter's
    hand-rolled Adam + decay bookkeeping vs a double-precision
    recomputation, and the NtsContext tape's graph-op/NN-segment
    interleaving with analytic gradients (cpp/host_unit_check.cpp)."""
    bin_ = os.path.join(REPO, "cpp", "build", "host_unit_check")
    if not os.path.exists(bin_):
        import __graft_entry__
        __graft_entry__._build_cpp()
    r = subprocess.run([bin_], capture_output=True, text=True, timeout=300,
                       env=_env())
    assert r.returncode == 0, f"host_unit_check:\n{r.stdout}\n{r.stderr}"
    assert "host_unit_check ok" in r.stdout
