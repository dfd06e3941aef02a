"""CPU-side checks of the C-ABI shared library: it builds, loads, and exports
every symbol include/dolomite_hip.h declares (no compute without a GPU)."""

import ctypes
import re
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
HEADER = REPO / "include" / "dolomite_hip.h"
SO = REPO / "dolomite_engine_amd" / "libdolomite_hip.so"


def _declared_symbols():
    text = HEADER.read_text()
    return sorted(set(re.findall(r"^int (dolomite_\w+)\(", text, flags=re.M)))


@pytest.fixture(scope="module")
def solib():
    if not SO.exists():
        from dolomite_engine_amd.csrc.build import build

        build()
    return ctypes.CDLL(str(SO))


def test_header_declares_expected_surface():
    syms = _declared_symbols()
    for required in [
        "dolomite_fa_varlen_fwd",
        "dolomite_fa_varlen_bwd",
        "dolomite_rmsnorm_fwd",
        "dolomite_rmsnorm_bwd",
        "dolomite_rope_qkv",
        "dolomite_ce_fwd",
        "dolomite_ce_bwd",
        "dolomite_adamw_step",
    ]:
        assert required in syms, required


def test_library_exports_every_declared_symbol(solib):
    for sym in _declared_symbols():
        assert hasattr(solib, sym), f"{sym} not exported by {SO.name}"


def test_abi_version(solib):
    fn = solib.dolomite_hip_abi_version
    fn.restype = ctypes.c_int32
    assert fn() == 2


def test_loader_signatures_cover_header():
    from dolomite_engine_amd.ops import hip

    declared = set(_declared_symbols())
    covered = set(hip._SIGNATURES)
    missing = declared - covered
    assert not missing, f"ops/hip.py signatures missing: {missing}"
