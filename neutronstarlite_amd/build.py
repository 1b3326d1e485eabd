"""In-tree build of the gfx950 HIP extension.

Builds neutronstarlite_amd/csrc/nts_hip.hip -> neutronstarlite_amd/libnts_hip.so
with hipcc --offload-arch=gfx950.  The built .so is git-ignored but travels
with the gpurun snapshot (it is NOT gpurun-ignored), so the GPU box uses the
binary compiled here.  hipcc cross-compiles without a GPU present.
"""
import os
import subprocess
import sys

_DIR = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(_DIR, "csrc", "nts_hip.hip")
SO = os.path.join(_DIR, "libnts_hip.so")
HIPCC = os.environ.get("HIPCC", "hipcc")


def needs_build():
    if not os.path.exists(SO):
        return True
    mt = os.path.getmtime
    hdr = os.path.join(_DIR, "..", "include", "nts_hip.h")
    return mt(SO) < max(mt(SRC), mt(hdr))


def build(verbose=True, force=False):
    if not force and not needs_build():
        return SO
    cmd = [
        HIPCC, "--offload-arch=gfx950", "-O3", "-std=c++17",
        "-fvisibility=default", "-shared", "-fPIC", SRC, "-o", SO,
    ]
    if verbose:
        print("+", " ".join(cmd), flush=True)
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        sys.stderr.write(r.stdout + r.stderr)
        raise RuntimeError("hipcc build failed")
    return SO


if __name__ == "__main__":
    build(force="--force" in sys.argv)
