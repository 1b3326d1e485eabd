"""The C-ABI library loads and exports every symbol include/nts_hip.h
declares (no compute calls — this container has no GPU)."""
import ctypes
import os
import re

import pytest

from tests.conftest import REPO

HDR = os.path.join(REPO, "include", "nts_hip.h")
SO = os.path.join(REPO, "neutronstarlite_amd", "libnts_hip.so")


def _declared_functions():
    text = open(HDR).read()
    text = re.sub(r"/\*.*?\*/", "", text, flags=re.S)
    names = re.findall(r"\b(nts_\w+)\s*\(", text)
    # drop type names
    return sorted(set(n for n in names if n != "nts_stream"))


def _built():
    if not os.path.exists(SO):
        import neutronstarlite_amd.build as b
        b.build()
    return SO


def test_header_symbols_all_exported():
    lib = ctypes.CDLL(_built())
    missing = [n for n in _declared_functions() if not hasattr(lib, n)]
    assert not missing, f"symbols declared in nts_hip.h but not exported: {missing}"
    assert len(_declared_functions()) >= 25


def test_shim_binds_and_reports_arch():
    import neutronstarlite_amd.shim as shim
    _built()
    l = shim.lib()
    assert l.nts_build_arch() == b"gfx950"


def test_shim_fails_loudly_when_missing(monkeypatch):
    import neutronstarlite_amd.shim as shim
    monkeypatch.setattr(shim, "_SO", "/nonexistent/libnts_hip.so")
    monkeypatch.setattr(shim, "_lib", None)
    with pytest.raises(shim.NtsHipMissing):
        shim.lib()


def test_so_is_gfx950_only():
    """The fat binary embeds exactly one offload arch: gfx950."""
    data = open(_built(), "rb").read()
    assert b"gfx950" in data
    for other in (b"gfx90a", b"gfx942", b"sm_80", b"sm_90"):
        assert other not in data
